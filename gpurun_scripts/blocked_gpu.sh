set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 700 python -m pytest tests/ -m gpu -x -q > gpurun_out/pytest_all_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_all_gpu.log
timeout 420 python bench.py --steps 10 --warmup 3 > gpurun_out/bench_rmat27_blocked.log 2>&1
echo "exit: $?" >> gpurun_out/bench_rmat27_blocked.log
timeout 240 python bench.py --scale 24 --edges $((1<<28)) --steps 10 --warmup 3 > gpurun_out/bench_rmat24_v2.log 2>&1
echo "exit: $?" >> gpurun_out/bench_rmat24_v2.log
timeout 420 python bench.py --app sssp --steps 3 --warmup 1 > gpurun_out/bench_sssp27.log 2>&1
echo "exit: $?" >> gpurun_out/bench_sssp27.log
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_pr27_v2 -- python /root/repo/bench.py --steps 10 --warmup 2 > /root/repo/gpurun_out/prof_pr27_v2.log 2>&1
echo "prof exit: $?" >> /root/repo/gpurun_out/prof_pr27_v2.log
