set -x
ulimit -c 0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof10k -o s10k -- python /root/repo/bench.py --bench-config burst10k --steps 20 --warmup 3 > /root/repo/gpurun_out/r02_final2_prof10k.log 2>&1
echo "prof10k rc=$?"
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof1m -o s1m -- python /root/repo/bench.py --bench-config burst1m --steps 8 --warmup 2 > /root/repo/gpurun_out/r02_final2_prof1m.log 2>&1
echo "prof1m rc=$?"
ls /tmp/prof10k /tmp/prof1m
cp /tmp/prof10k/*kernel_stats.csv /root/repo/gpurun_out/r02_final2_10k_kernel_stats.csv 2>/dev/null
cp /tmp/prof1m/*kernel_stats.csv /root/repo/gpurun_out/r02_final2_1m_kernel_stats.csv 2>/dev/null
head -8 /root/repo/gpurun_out/r02_final2_10k_kernel_stats.csv
