set -x
cd /root/repo
mkdir -p gpurun_out
{
export LUX_SYNC_CHECK=1 AMD_SERIALIZE_KERNEL=3
timeout 300 python - <<'PYEOF'
import torch, numpy as np
from lux_amd.engine import DeviceCSC, GraphPart
from lux_amd.cf_engine import CFEngine
from lux_amd.graph import Graph
from lux_amd import cpu_ref
K = 16
full = DeviceCSC.bipartite(500, 128, 20000, seed=3)
part = GraphPart(full, 1, 0)
part.build_bins()
print("bins:", part.n0, part.n1, part.n2, part.nbig, flush=True)
eng = CFEngine(part, K=K)
eng.step()
torch.cuda.synchronize()
got = eng.vectors().cpu().numpy()
g = Graph.bipartite(500, 128, 20000, seed=3)
want = cpu_ref.cf(g, K, 1)
bad = np.where(np.abs(got - want).max(axis=1) > 1e-6)[0]
deg = np.diff(np.concatenate([[0], g.col_end])).astype(int)
print("mismatched vertices:", len(bad), flush=True)
print("their degrees:", sorted(deg[bad].tolist())[:50])
print("degree histogram of mismatches: <32:", (deg[bad]<32).sum(),
      "32..2048:", ((deg[bad]>=32)&(deg[bad]<2048)).sum(),
      ">=2048:", (deg[bad]>=2048).sum())
if len(bad):
    v = int(bad[0])
    print("example v:", v, "deg:", deg[v], "got:", got[v][:4], "want:", want[v][:4])
# second iteration to see if context survives
eng.step(); torch.cuda.synchronize()
print("context alive after 2 steps", flush=True)
PYEOF
echo "debug exit: $?"
} > gpurun_out/debug_cf2.log 2>&1
unset LUX_SYNC_CHECK AMD_SERIALIZE_KERNEL
timeout 300 python -m pytest tests/test_gpu_pull.py -q > gpurun_out/pytest_pull_fresh.log 2>&1
echo "pull exit: $?" >> gpurun_out/pytest_pull_fresh.log
timeout 300 python -m pytest tests/test_gpu_push.py -q > gpurun_out/pytest_push_fresh.log 2>&1
echo "push exit: $?" >> gpurun_out/pytest_push_fresh.log
timeout 300 python -m pytest tests/test_gpu_cf.py -q > gpurun_out/pytest_cf_fresh.log 2>&1
echo "cf exit: $?" >> gpurun_out/pytest_cf_fresh.log
tail -12 gpurun_out/debug_cf2.log
tail -4 gpurun_out/pytest_pull_fresh.log gpurun_out/pytest_push_fresh.log gpurun_out/pytest_cf_fresh.log
