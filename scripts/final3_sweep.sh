set -x
ulimit -c 0
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02_final3_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r02_final3_pytest.log
FUZZ_ITERS=200 timeout 900 python scripts/fuzz_gpu.py > gpurun_out/r02_final3_fuzz.log 2>&1
echo "fuzz rc=$?"; tail -1 gpurun_out/r02_final3_fuzz.log
for cfg in burst10k gpu1k maintain500; do
  timeout 240 python bench.py --bench-config $cfg --steps 30 --warmup 5 > gpurun_out/r02_final3_$cfg.json 2>/dev/null
  python -c "import json; d=json.load(open('gpurun_out/r02_final3_$cfg.json')); print('$cfg p50=%.3f value=%.0f' % (d['config'].get('decision_p50_ms', d['ms_per_step']), d['value']))"
done
timeout 300 python bench.py --bench-config burst1m --steps 10 --warmup 2 > gpurun_out/r02_final3_burst1m.json 2>/dev/null
python -c "import json; d=json.load(open('gpurun_out/r02_final3_burst1m.json')); print('burst1m p50=%.3f value=%.0f' % (d['config']['decision_p50_ms'], d['value']))"
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke(); print('smoke ok')"
