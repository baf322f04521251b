set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -2 > gpurun_out/mid_pytest.log
{
echo "=== pagerank shift 25 (default) ==="
timeout 300 python bench.py --steps 6 --warmup 1 2>/dev/null
echo "=== pagerank shift 24 ==="
LUX_BLOCK_SHIFT=24 timeout 300 python bench.py --steps 6 --warmup 1 2>/dev/null
echo "=== pagerank shift 26 ==="
LUX_BLOCK_SHIFT=26 timeout 300 python bench.py --steps 6 --warmup 1 2>/dev/null
} > gpurun_out/pr_shift.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_cc -o cc -- python /root/repo/bench.py --app cc --steps 5 --warmup 1 > /dev/null 2>&1
timeout 240 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_sssp2 -o sssp2 -- python /root/repo/bench.py --app sssp --steps 2 --warmup 1 > /dev/null 2>&1
cat /root/repo/gpurun_out/mid_pytest.log
grep -E "===|GTEPS" /root/repo/gpurun_out/pr_shift.log
