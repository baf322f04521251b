set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 900 python -m pytest tests/ -m gpu -q > gpurun_out/pytest_v3.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_v3.log
timeout 420 python bench.py --steps 10 --warmup 3 > gpurun_out/bench27_v3.log 2>&1
echo "exit: $?" >> gpurun_out/bench27_v3.log
timeout 240 python bench.py --scale 24 --edges $((1<<28)) --steps 10 --warmup 3 > gpurun_out/bench24_v3.log 2>&1
echo "exit: $?" >> gpurun_out/bench24_v3.log
timeout 300 python bench.py --app cf --steps 5 --warmup 2 > gpurun_out/bench_cf_v3.log 2>&1
echo "exit: $?" >> gpurun_out/bench_cf_v3.log
timeout 400 python bench.py --app cc --steps 2 --warmup 1 > gpurun_out/bench_cc_v3.log 2>&1
echo "exit: $?" >> gpurun_out/bench_cc_v3.log
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_v3 -- python /root/repo/bench.py --steps 10 --warmup 2 > /root/repo/gpurun_out/prof_v3.log 2>&1
echo "prof exit: $?" >> /root/repo/gpurun_out/prof_v3.log
tail -4 gpurun_out/pytest_v3.log /root/repo/gpurun_out/bench27_v3.log
