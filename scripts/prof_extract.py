"""Extract per-kernel dispatch stats from a rocprofv3 results .db."""
import sqlite3, sys

db, out = sys.argv[1], sys.argv[2]
con = sqlite3.connect(db)
q = """SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1000.0,
              AVG(d.end-d.start)/1000.0
       FROM kernel_dispatch d JOIN kernel_symbol s ON d.kernel_id=s.id
       GROUP BY s.display_name ORDER BY 3 DESC LIMIT 20"""
with open(out, "w") as f:
    f.write(f"{'kernel':<64} {'calls':>7} {'total_us':>12} {'avg_us':>10}\n")
    for name, n, tot, avg in con.execute(q):
        f.write(f"{name[:64]:<64} {n:>7} {tot:>12.1f} {avg:>10.2f}\n")
print(open(out).read())
