set -x
cd /tmp && export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
# full suite, one process, twice (stability check after test fix)
cd /root/repo
timeout 600 python -m pytest tests/ -m gpu -q > gpurun_out/pytest_full_a.log 2>&1
echo "exit: $?" >> gpurun_out/pytest_full_a.log
timeout 600 python -m pytest tests/ -m gpu -q > gpurun_out/pytest_full_b.log 2>&1
echo "exit: $?" >> gpurun_out/pytest_full_b.log
cd /tmp
# PMC counters for the pull iteration (own run, counters only + kernel trace)
timeout 420 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE L2CacheHit MemUnitStalled --kernel-trace -d /root/repo/gpurun_out/pmc_pr27 -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/pmc_pr27.log 2>&1
echo "pmc exit: $?" >> /root/repo/gpurun_out/pmc_pr27.log
tail -3 /root/repo/gpurun_out/pytest_full_a.log
tail -3 /root/repo/gpurun_out/pytest_full_b.log
tail -6 /root/repo/gpurun_out/pmc_pr27.log
