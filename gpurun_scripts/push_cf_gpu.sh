set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 700 python -m pytest tests/test_gpu_push.py tests/test_gpu_cf.py -x -q > gpurun_out/pytest_push_cf.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_push_cf.log
timeout 240 python bench.py --app sssp --scale 24 --edges $((1<<28)) --steps 3 --warmup 1 > gpurun_out/bench_sssp24.log 2>&1
echo "exit: $?" >> gpurun_out/bench_sssp24.log
timeout 300 python bench.py --app cf --steps 5 --warmup 2 > gpurun_out/bench_cf.log 2>&1
echo "exit: $?" >> gpurun_out/bench_cf.log
timeout 400 python bench.py --app cc --steps 2 --warmup 1 > gpurun_out/bench_cc.log 2>&1
echo "exit: $?" >> gpurun_out/bench_cc.log
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_pr27 -- python /root/repo/bench.py --steps 10 --warmup 2 > /root/repo/gpurun_out/prof_pr27.log 2>&1
echo "prof exit: $?" >> /root/repo/gpurun_out/prof_pr27.log
