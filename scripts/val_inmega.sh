set -x
ulimit -c 0
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02_inmega_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/r02_inmega_pytest.log
for cfg in burst10k gpu1k; do
  timeout 240 python bench.py --bench-config $cfg --steps 30 --warmup 5 > gpurun_out/r02_inmega_$cfg.json 2>gpurun_out/r02_inmega_$cfg.err
  echo "rc=$? cfg=$cfg"
done
timeout 300 python bench.py --bench-config burst1m --steps 10 --warmup 2 > gpurun_out/r02_inmega_burst1m.json 2>gpurun_out/r02_inmega_burst1m.err
echo "rc=$? cfg=burst1m"
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof -o inmega -- python /root/repo/bench.py --bench-config burst10k --steps 20 --warmup 3 > /root/repo/gpurun_out/r02_inmega_prof.log 2>&1
echo "prof rc=$?" >> /root/repo/gpurun_out/r02_inmega_prof.log
cp /tmp/prof/*inmega*stats* /root/repo/gpurun_out/ 2>/dev/null || ls /tmp/prof >> /root/repo/gpurun_out/r02_inmega_prof.log
tail -4 /root/repo/gpurun_out/r02_inmega_pytest.log
grep -h '"metric"' /root/repo/gpurun_out/r02_inmega_*.json
