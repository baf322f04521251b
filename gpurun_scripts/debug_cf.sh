set -x
cd /root/repo
mkdir -p gpurun_out
export LUX_SYNC_CHECK=1
timeout 200 python - > gpurun_out/debug_cf.log 2>&1 <<'PYEOF'
import torch, numpy as np
from lux_amd.engine import DeviceCSC, GraphPart
from lux_amd import _native_gpu as ng
full = DeviceCSC.bipartite(500, 128, 20000, seed=3)
part = GraphPart(full, 1, 0)
part.build_bins()
print("bins:", part.n0, part.n1, part.n2, part.nbig, flush=True)
print("row_ptr tail:", part.row_ptr[-5:].cpu().tolist(), "ep:", part.ep)
print("weight[:5]:", part.weight[:5].cpu().tolist())
from lux_amd.cf_engine import CFEngine
eng = CFEngine(part, K=16)
print("old[:4]:", eng.old[:4].cpu().tolist())
eng.step()
torch.cuda.synchronize()
print("new_part[:4]:", eng.new_part[:4].cpu().tolist())
print("new_part nonzero:", (eng.new_part != 0).sum().item(), "/", eng.new_part.numel())
print("item vec (vertex 500):", eng.vectors()[500][:4].cpu().tolist())
PYEOF
echo "debug exit: $?" >> gpurun_out/debug_cf.log
unset LUX_SYNC_CHECK
timeout 900 python -m pytest tests/ -m gpu -q > gpurun_out/pytest_all2.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_all2.log
tail -20 gpurun_out/debug_cf.log
tail -30 gpurun_out/pytest_all2.log
