set -x
cd /root/repo
mkdir -p gpurun_out
python - << 'PYEOF' > gpurun_out/cc_breakdown.log 2>&1
import time
import torch
from lux_amd.engine import DeviceCSC, GraphPart
from lux_amd.push_engine import PushEngine

full = DeviceCSC.rmat_folded(41652230, 1468365182, seed=1)
part = GraphPart(full, 1, 0)
eng = PushEngine(part, PushEngine.MODE_MAX)
# warm run
eng.run(); eng.reset()
torch.cuda.synchronize()
t0 = time.perf_counter()
it = 0
while True:
    s = time.perf_counter()
    total = eng.step()
    torch.cuda.synchronize()
    e = time.perf_counter()
    st = eng.stats[-1]
    print(f"iter {it:3d}: {1000*(e-s):8.2f} ms  old_fq={st['old_frontier']:>10} "
          f"pull={st['pull_fallback']} out_dense={st['out_dense']} my_new={st['my_new']}")
    it += 1
    if total == 0 or it > 80:
        break
print(f"TOTAL {1000*(time.perf_counter()-t0):.1f} ms in {it} iters")
PYEOF
cat gpurun_out/cc_breakdown.log
