set -x
cd /root/repo
mkdir -p gpurun_out
{
check() {
python - <<PYEOF
import sys
$1
import torch
t = torch.zeros(256, dtype=torch.int32, device="cuda")
from lux_amd import _native_gpu as ng
ng.rmat_edges(torch.cuda.current_stream().cuda_stream, 1, 8, 256, t, t)
torch.cuda.synchronize()
ok = (t.cpu().numpy() != 0).any()
print("IMPORTS=[$1] kernel_ran=", bool(ok))
PYEOF
}
check ""
check "import pytest"
check "import tests.conftest"
check "import tests.test_cpu_engines"
check "import tests.test_dist_cpu"
check "import tests.test_checkpoint"
check "import tests.test_luxio, tests.test_partition, tests.test_rmat"
} > gpurun_out/bisect2.log 2>&1
timeout 200 python -m pytest "tests/test_gpu_cf.py::test_cf_vs_cpu[16]" -q > gpurun_out/repro_a.log 2>&1
echo "exit $?" >> gpurun_out/repro_a.log
PYTEST_DISABLE_PLUGIN_AUTOLOAD=1 timeout 200 python -m pytest "tests/test_gpu_cf.py::test_cf_vs_cpu[16]" -q > gpurun_out/repro_b.log 2>&1
echo "exit $?" >> gpurun_out/repro_b.log
grep -a "IMPORTS" gpurun_out/bisect2.log
tail -a -3 gpurun_out/repro_a.log
tail -a -3 gpurun_out/repro_b.log
