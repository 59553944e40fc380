set -x
ulimit -c 0
mkdir -p gpurun_out
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke(); print('smoke ok')" > gpurun_out/r02_rehearsal_smoke.log 2>&1
echo "smoke rc=$?"
timeout 600 python bench.py > gpurun_out/r02_rehearsal_bench_default.json 2>gpurun_out/r02_rehearsal_bench_default.err
echo "bench-default rc=$?"
timeout 300 python bench.py --bench-config burst10m --steps 5 --warmup 1 > gpurun_out/r02_rehearsal_10m.json 2>/dev/null
echo "10m rc=$?"
CYCLES=500 timeout 900 python scripts/bench_watch_resident.py > gpurun_out/r02_rehearsal_soak500.log 2>&1
echo "soak rc=$?"
tail -2 gpurun_out/r02_rehearsal_smoke.log
grep -h '"metric"' gpurun_out/r02_rehearsal_bench_default.json gpurun_out/r02_rehearsal_10m.json | cut -c1-200
tail -4 gpurun_out/r02_rehearsal_soak500.log
