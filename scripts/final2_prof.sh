set -x
ulimit -c 0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof10k -o s10k -- python /root/repo/bench.py --bench-config burst10k --steps 20 --warmup 3 > /root/repo/gpurun_out/r02_final2_prof10k.log 2>&1
echo "prof10k rc=$?"
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof1m -o s1m -- python /root/repo/bench.py --bench-config burst1m --steps 8 --warmup 2 > /root/repo/gpurun_out/r02_final2_prof1m.log 2>&1
echo "prof1m rc=$?"
python /root/repo/scripts/prof_extract.py /tmp/prof10k/s10k_results.db /root/repo/gpurun_out/r02_final2_prof10k_kernels.txt
python /root/repo/scripts/prof_extract.py /tmp/prof1m/s1m_results.db /root/repo/gpurun_out/r02_final2_prof1m_kernels.txt
