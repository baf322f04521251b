set -x
cd /root/repo
mkdir -p gpurun_out
rocm-smi --showproductname 2>&1 | head -5 > gpurun_out/gpuinfo.txt
timeout 420 python -m pytest tests/test_gpu_pull.py -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.log
timeout 180 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke.log 2>&1
echo "smoke exit: $?" >> gpurun_out/smoke.log
timeout 240 python bench.py --scale 24 --edges $((1<<28)) --steps 10 --warmup 3 > gpurun_out/bench_rmat24.log 2>&1
echo "bench24 exit: $?" >> gpurun_out/bench_rmat24.log
timeout 420 python bench.py --steps 10 --warmup 3 > gpurun_out/bench_rmat27.log 2>&1
echo "bench27 exit: $?" >> gpurun_out/bench_rmat27.log
tail -3 gpurun_out/pytest_gpu.log gpurun_out/smoke.log gpurun_out/bench_rmat24.log gpurun_out/bench_rmat27.log
