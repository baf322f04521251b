set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.log
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_cf -o cf -- python /root/repo/bench.py --app cf --steps 10 --warmup 2 > /root/repo/gpurun_out/prof_cf.log 2>&1
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_cf_als -o cfals -- python /root/repo/bench.py --app cf_als --steps 5 --warmup 1 > /root/repo/gpurun_out/prof_cf_als.log 2>&1
timeout 700 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE TCC_HIT TCC_MISS --kernel-trace -d /root/repo/gpurun_out/pmc_pr -o pmcpr -- python /root/repo/bench.py --scale 26 --edges $((1<<30)) --steps 2 --warmup 1 > /root/repo/gpurun_out/pmc_pr.log 2>&1
echo "pmc exit: $?" >> /root/repo/gpurun_out/pmc_pr.log
tail -3 /root/repo/gpurun_out/pytest_gpu.log
grep -hE "GTEPS" /root/repo/gpurun_out/prof_cf.log /root/repo/gpurun_out/prof_cf_als.log || true
find /root/repo/gpurun_out/prof_cf /root/repo/gpurun_out/pmc_pr -name "*.csv" | head
