set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests -m gpu -q 2>&1 | tail -2 > gpurun_out/sssp_v3.log
bash gpurun_scripts/sssp_breakdown.sh > /dev/null 2>&1
cat gpurun_out/sssp_breakdown.log >> gpurun_out/sssp_v3.log
{ echo "=== sssp bench ==="; timeout 300 python bench.py --app sssp --steps 3 --warmup 1 2>/dev/null; } >> gpurun_out/sssp_v3.log
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_sssp -o sssp -- python /root/repo/bench.py --app sssp --steps 2 --warmup 1 > /dev/null 2>&1
cat /root/repo/gpurun_out/sssp_v3.log
