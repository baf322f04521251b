set -x
cd /root/repo
mkdir -p gpurun_out
python - << 'PYEOF' > gpurun_out/stability.log 2>&1
import numpy as np
import torch
from lux_amd.engine import DeviceCSC, GraphPart, PagerankEngine
from lux_amd.push_engine import PushEngine
from lux_amd import cpu_ref
from lux_amd.graph import Graph

# 100-iteration PR: finite, stable sum, no leak
full = DeviceCSC.rmat(22, 1 << 26, seed=3)
eng = PagerankEngine(GraphPart(full, 1, 0))
m0 = torch.cuda.memory_allocated()
for i in range(100):
    eng.step()
r = eng.ranks()
m1 = torch.cuda.memory_allocated()
undiv = r * eng.deg.clamp(min=1).to(torch.float32)
print(f"PR100: finite={bool(torch.isfinite(r).all())} "
      f"sum={float(undiv.sum()):.4f} mem_delta={m1-m0}")

# SSSP from 20 random sources with the check oracle
full = DeviceCSC.rmat(20, 1 << 23, seed=5)
part = GraphPart(full, 1, 0)
eng = PushEngine(part, PushEngine.MODE_MIN, source=0)
bad = 0
rng = np.random.default_rng(1)
for src in rng.integers(0, 1 << 20, 20):
    eng.reset(int(src))
    eng.run()
    bad += eng.check()
print(f"SSSP20: total oracle violations = {bad}")
PYEOF
cat gpurun_out/stability.log
