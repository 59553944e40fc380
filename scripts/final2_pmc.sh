set -x
ulimit -c 0
mkdir -p gpurun_out
timeout 180 python scripts/probe_decide_host.py > gpurun_out/r02_final2_hostprobe.log 2>&1
echo "probe rc=$?"
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT SQ_WAVES --output-format csv -d /tmp/pmc10k -o pmc10k -- python /root/repo/bench.py --bench-config burst10k --steps 10 --warmup 3 > /root/repo/gpurun_out/r02_final2_pmc.log 2>&1
echo "pmc rc=$?"
ls /tmp/pmc10k
cp /tmp/pmc10k/*counter*csv /tmp/pmc10k/*pmc*csv /root/repo/gpurun_out/ 2>/dev/null
tail -25 /root/repo/gpurun_out/r02_final2_hostprobe.log
