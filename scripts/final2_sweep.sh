set -x
ulimit -c 0
mkdir -p gpurun_out
FUZZ_ITERS=300 timeout 1200 python scripts/fuzz_gpu.py > gpurun_out/r02_final2_fuzz300.log 2>&1
echo "rc=$?" >> gpurun_out/r02_final2_fuzz300.log
timeout 240 python bench.py --bench-config maintain500 --steps 30 --warmup 5 > gpurun_out/r02_final2_maintain500.json 2>/dev/null; echo "m500 rc=$?"
timeout 300 python bench.py --bench-config e2e10k --steps 10 --warmup 2 > gpurun_out/r02_final2_e2e10k.json 2>/dev/null; echo "e2e rc=$?"
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof10k -o s10k -- python /root/repo/bench.py --bench-config burst10k --steps 20 --warmup 3 > /root/repo/gpurun_out/r02_final2_prof10k.log 2>&1
echo "prof10k rc=$?"
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof1m -o s1m -- python /root/repo/bench.py --bench-config burst1m --steps 8 --warmup 2 > /root/repo/gpurun_out/r02_final2_prof1m.log 2>&1
echo "prof1m rc=$?"
for d in prof10k prof1m; do
  db=$(ls /tmp/$d/*.db 2>/dev/null | head -1)
  [ -n "$db" ] && sqlite3 "$db" "SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1000.0, AVG(d.end-d.start)/1000.0 FROM kernel_dispatch d JOIN kernel_symbol s ON d.kernel_id=s.id GROUP BY s.display_name ORDER BY 3 DESC LIMIT 14;" > /root/repo/gpurun_out/r02_final2_${d}_kernels.txt 2>&1
done
tail -3 /root/repo/gpurun_out/r02_final2_fuzz300.log
grep -h '"metric"' /root/repo/gpurun_out/r02_final2_*.json | cut -c1-220
head -6 /root/repo/gpurun_out/r02_final2_prof10k_kernels.txt
