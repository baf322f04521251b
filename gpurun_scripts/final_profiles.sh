set -x
cd /tmp && export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_pr -o pr -- python /root/repo/bench.py --steps 6 --warmup 1 > /root/repo/gpurun_out/prof_pr.log 2>&1
timeout 240 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_ccuf -o ccuf -- python /root/repo/bench.py --app cc --steps 5 --warmup 1 > /root/repo/gpurun_out/prof_ccuf.log 2>&1
timeout 240 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_cf32 -o cf32 -- python /root/repo/bench.py --app cf --steps 8 --warmup 1 > /root/repo/gpurun_out/prof_cf32.log 2>&1
timeout 420 rocprofv3 --pmc FETCH_SIZE --kernel-trace -d /root/repo/gpurun_out/pmc_fetch -o fetch -- python /root/repo/bench.py --scale 26 --edges $((1<<30)) --steps 2 --warmup 0 > /root/repo/gpurun_out/pmc_fetch.log 2>&1
echo "pmc exit $?"
grep -h ms_per_step /root/repo/gpurun_out/prof_pr.log /root/repo/gpurun_out/prof_ccuf.log /root/repo/gpurun_out/prof_cf32.log 2>/dev/null | head -3
ls /root/repo/gpurun_out/prof_pr /root/repo/gpurun_out/pmc_fetch 2>/dev/null
