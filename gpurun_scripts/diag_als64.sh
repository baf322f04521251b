set -x
cd /root/repo
python - << 'PYEOF' > gpurun_out/diag_als64.log 2>&1
import numpy as np
import torch
from lux_amd import cpu_ref
from lux_amd.cf_engine import CFALSEngine
from lux_amd.engine import DeviceCSC, GraphPart
from lux_amd.graph import Graph

K = 64
nu, ni, ne = 400, 100, 15000
full = DeviceCSC.bipartite(nu, ni, ne, seed=11)
part = GraphPart(full, 1, 0)
eng = CFALSEngine(part, K=K)
rng = np.random.default_rng(42)
init = (rng.standard_normal((part.nv, K)) * 0.1 + 0.1).astype(np.float32)
eng.old.copy_(torch.from_numpy(init.ravel()))
eng.step()
got = eng.vectors().cpu().numpy()
g = Graph.bipartite(nu, ni, ne, seed=11)
want = cpu_ref.cf_als(g, K, 1, init=init)
deg = np.diff(np.concatenate([[0], g.col_end])).astype(int)
err = np.abs(got - want)
tol = 3e-4 + 3e-3 * np.abs(want)
bad = np.nonzero((err > tol).any(axis=1))[0]
print("bad vertices:", bad)
for v in bad[:10]:
    nb = int((err[v] > tol[v]).sum())
    print(f"v={v} deg={deg[v]} bad_dims={nb} maxabs={err[v].max():.5f} "
          f"dims={np.nonzero(err[v] > tol[v])[0][:12]}")
# bin membership
part.build_bins()
b0 = part.bin0[:part.n0].cpu().numpy(); b1 = part.bin1[:part.n1].cpu().numpy()
b2v = part.bin2v[:part.nbig].cpu().numpy()
for v in bad[:10]:
    loc = "bin0" if v in b0 else "bin1" if v in b1 else "hub" if v in b2v else "??"
    print(f"v={v}: {loc}")
# recompute the worst vertex's solve in f64 from the SAME init to isolate
v = int(bad[0])
b = int(g.col_end[v-1]) if v > 0 else 0
e = int(g.col_end[v])
S = init[g.src[b:e]].astype(np.float64)
w = g.weight[b:e].astype(np.float64)
G = S.T @ S + 0.001 * np.eye(K)
d = np.linalg.solve(G, S.T @ w)
print("cond(G) =", np.linalg.cond(G))
print("ref-vs-want match:", np.allclose(d, want[v], atol=1e-5))
print("got[v][:8] =", got[v][:8])
print("want[v][:8]=", want[v][:8])
PYEOF
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.log
timeout 240 python bench.py --app cf_als --steps 5 --warmup 1 > gpurun_out/als_bench2.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --kernel-trace -d /root/repo/gpurun_out/pmc_fs -o fs -- python /root/repo/bench.py --scale 26 --edges $((1<<30)) --steps 2 --warmup 0 > /root/repo/gpurun_out/pmc_fs.log 2>&1
timeout 500 rocprofv3 --pmc TCC_HIT TCC_MISS --kernel-trace -d /root/repo/gpurun_out/pmc_tcc -o tcc -- python /root/repo/bench.py --scale 26 --edges $((1<<30)) --steps 2 --warmup 0 > /root/repo/gpurun_out/pmc_tcc.log 2>&1
cat /root/repo/gpurun_out/diag_als64.log
tail -3 /root/repo/gpurun_out/pytest_gpu.log
grep -h GTEPS /root/repo/gpurun_out/als_bench2.log
ls /root/repo/gpurun_out/pmc_fs /root/repo/gpurun_out/pmc_tcc 2>/dev/null
