set -x
ulimit -c 0
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_gpu_engine.py tests/test_differential.py -m gpu -q > gpurun_out/r02_single_pytest.log 2>&1
echo "pytest rc=$?"
tail -3 gpurun_out/r02_single_pytest.log
for mp in 0 2048; do
  MI355X_AS_SINGLE_MAX_P=$mp timeout 240 python bench.py --bench-config gpu1k --steps 40 --warmup 8 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('gpu1k single_max_p=$mp p50=%.3f ms=%.3f' % (d['config']['decision_p50_ms'], d['ms_per_step']))"
done
for mp in 0 16384; do
  MI355X_AS_SINGLE_MAX_P=$mp timeout 240 python bench.py --bench-config burst10k --steps 30 --warmup 5 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('burst10k single_max_p=$mp p50=%.3f ms=%.3f' % (d['config']['decision_p50_ms'], d['ms_per_step']))"
done
FUZZ_ITERS=120 timeout 900 python scripts/fuzz_gpu.py > gpurun_out/r02_single_fuzz.log 2>&1
echo "fuzz rc=$?"; tail -2 gpurun_out/r02_single_fuzz.log
