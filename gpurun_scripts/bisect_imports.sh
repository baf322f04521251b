set -x
cd /root/repo
mkdir -p gpurun_out
{
check() {
python - <<PYEOF
import sys
$1
import torch
t = torch.empty(256, dtype=torch.int32, device="cuda")
from lux_amd import _native_gpu as ng
ng.rmat_edges(torch.cuda.current_stream().cuda_stream, 1, 8, 256, t, t)
torch.cuda.synchronize()
ok = (t.cpu().numpy() != 0).any()
print("IMPORTS=[$1] kernel_ran=", bool(ok))
PYEOF
}
check ""
check "import tests.conftest"
check "import tests.test_dist_cpu"
check "import tests.test_cpu_engines"
check "import tests.test_luxio, tests.test_partition, tests.test_rmat, tests.test_checkpoint"
check "import torch.multiprocessing"
check "import torch.distributed"
} > gpurun_out/bisect.log 2>&1
timeout 300 python -m pytest tests/test_gpu_cf.py -m gpu -q > gpurun_out/cf_mgpu.log 2>&1
echo "cf -m gpu exit: $?" >> gpurun_out/cf_mgpu.log
grep -E "IMPORTS|kernel_ran|Error|error" gpurun_out/bisect.log | head -30
tail -2 gpurun_out/cf_mgpu.log
