"""Distributed collaborative-filtering engine (pull model, K-dim latent
vectors). Exchange pattern identical to PageRank (all-gather(v) of vector
slices); compute per sweep is src-vector gather bound (256 B per edge at
K=64)."""
import math

import torch

from . import _native_gpu as ng
from . import dist as dx
from .engine import GraphPart

F32 = torch.float32


def _stream():
    return torch.cuda.current_stream().cuda_stream


def als_init(nv, K):
    """Deterministic jittered ALS init: sqrt(1/K) * [0.5, 1.5) per
    component (splitmix64 hash of the flat index; identical on every
    rank). The reference's constant sqrt(1/K) init makes the FIRST
    alternating half-sweep rank-1 degenerate — every user lands on the
    same line, the item solve then amplifies null-space noise by 1/lambda
    (measured: max|item| ~270 after one fp32 sweep, divergent feedback in
    bf16) — so the exact-solve optimizer needs the degeneracy broken at
    init. SGD keeps the reference's constant init."""
    import math
    import numpy as np
    i = np.arange(nv * K, dtype=np.uint64)
    z = (i + np.uint64(0x9E3779B97F4A7C15)) * np.uint64(0xBF58476D1CE4E5B9)
    z ^= z >> np.uint64(30)
    z *= np.uint64(0x94D049BB133111EB)
    z ^= z >> np.uint64(27)
    u = (z >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    return (math.sqrt(1.0 / K) * (0.5 + u)).astype(np.float32)


class CFEngine:
    def __init__(self, part: GraphPart, K=64):
        assert part.weight is not None, "CF needs a weighted graph"
        assert K <= 256
        self.part = part
        self.K = K
        part.build_bins()
        device = part.device
        # reference init: every component sqrt(1/K) (colfilter_gpu.cu:260-264)
        v0 = math.sqrt(1.0 / K)
        self.old = torch.full((part.nv * K,), v0, dtype=F32, device=device)
        self.new_part = torch.empty(part.vp * K, dtype=F32, device=device)
        self.verts_elems = [v * K for v in part.verts_all]
        self.left_elems = [l * K for l in part.row_left_all]

    def step(self):
        p = self.part
        # seed: old*(1 - GAMMA*LAMBDA); sweeps add GAMMA*acc (see cf.hip)
        self.new_part.copy_(
            self.old.narrow(0, p.row_left * self.K, p.vp * self.K))
        self.new_part.mul_(1.0 - 0.00000035 * 0.001)
        ng.cf_iter(_stream(), p.n0, p.bin0, p.n1, p.bin1, p.n2, p.bin2,
                   p.nbig, p.bin2v, p.row_ptr, p.col, p.weight, self.old,
                   self.new_part, p.row_left, self.K)
        dx.all_gather_slices(self.old, self.new_part, self.verts_elems,
                             self.left_elems, my_index=p.p)

    def vectors(self):
        return self.old.view(self.part.nv, self.K)


class CFALSEngine:
    """ALS sweeps via MFMA Gram accumulation + per-wave Cholesky
    (src/gpu/cf_als.hip). Same data movement and exchange as CFEngine;
    an alternative optimizer that reaches the SGD fixed point in far fewer
    sweeps. K <= 64 (the MFMA tile grid is 4x4 of 16x16).

    When the partition knows the bipartite user/item boundary
    (part.n_users), each sweep is true Gauss-Seidel ALTERNATION: solve all
    user rows against the old item factors, publish, then solve item rows
    against the UPDATED users. In-place alternation is exact here because
    a bipartite row only ever reads factors from the other side. The bin
    lists are split once at init (users first); no kernel changes — the
    item half-sweep passes offset bin/gram pointers. Without n_users
    (e.g. a graph loaded from .lux) the sweep falls back to the
    simultaneous-Jacobi update of every row from old values."""

    def __init__(self, part: GraphPart, K=64):
        assert part.weight is not None, "CF needs a weighted graph"
        assert K <= 64, "ALS MFMA path covers K <= 64"
        self.part = part
        self.K = K
        part.build_bins()
        device = part.device
        self.old = torch.from_numpy(als_init(part.nv, K)).to(device)
        self.new_part = torch.empty(part.vp * K, dtype=F32, device=device)
        self.verts_elems = [v * K for v in part.verts_all]
        self.left_elems = [l * K for l in part.row_left_all]
        nu = part.n_users
        self.lb = None if nu is None else min(max(nu - part.row_left, 0),
                                              part.vp)
        if self.lb is not None:
            # stable-partition every bin list: local user rows first
            def _split(t, n, key):
                t = t[:n]
                m = key(t)
                return torch.cat([t[m], t[~m]]), int(m.sum())
            lb = self.lb
            self.bin0s, self.n0u = _split(part.bin0, part.n0,
                                          lambda t: t < lb)
            self.bin1s, self.n1u = _split(part.bin1, part.n1,
                                          lambda t: t < lb)
            b2 = part.bin2[:part.n2 * 2].view(-1, 2)
            m2 = b2[:, 0] < lb
            self.bin2s = torch.cat([b2[m2], b2[~m2]]).reshape(-1)
            self.n2u = int(m2.sum())
            self.bin2vs, self.nbigu = _split(part.bin2v, part.nbig,
                                             lambda t: t < lb)
        if part.nbig:
            self.hubidx = torch.full((part.vp,), -1, dtype=torch.int32,
                                     device=device)
            if self.lb is None:
                hubs = part.bin2v[:part.nbig].long()
                self.hubidx[hubs] = torch.arange(
                    part.nbig, dtype=torch.int32, device=device)
            else:
                # phase-RELATIVE scratch slots: the item half-sweep gets
                # gram/rhs pointers offset by nbigu, so item hub i maps
                # to slot (i - nbigu) of the offset base
                slot = torch.arange(part.nbig, dtype=torch.int32,
                                    device=device)
                slot[self.nbigu:] -= self.nbigu
                self.hubidx[self.bin2vs.long()] = slot
            self.gram = torch.empty(part.nbig * 64 * 64, dtype=F32,
                                    device=device)
            self.rhs_h = torch.empty(part.nbig * 64, dtype=F32,
                                     device=device)
        else:
            self.hubidx = self.gram = self.rhs_h = None
        # bf16 gather replica: MEASURED SLOWER both ways and off by
        # default (BENCHLOG r2.4/r2.5) — per-lane sub-dword gathers run
        # 2.5x below dword rate, and the dword-pair staging variant loses
        # the register-prefetch pipeline (20.4 vs 10.5 ms/sweep). The
        # fp32-gather + bf16-LDS-stage split is the winning shape.
        # LUX_ALS_BF_GATHER=1 re-enables the experiment (needs even K).
        import os
        self.old_bf = torch.empty(part.nv * K, dtype=torch.bfloat16,
                                  device=device) \
            if os.environ.get("LUX_ALS_BF_GATHER") == "1" \
            and K % 2 == 0 and not os.environ.get("LUX_ALS_F32") else None

    def _begin_sweep(self):
        p = self.part
        # seed with old slice: vertices with no in-edges keep their vector
        self.new_part.copy_(
            self.old.narrow(0, p.row_left * self.K, p.vp * self.K))
        if p.nbig:
            self.gram.zero_()
            self.rhs_h.zero_()

    def _publish(self):
        p = self.part
        dx.all_gather_slices(self.old, self.new_part, self.verts_elems,
                             self.left_elems, my_index=p.p)

    def half_step(self, phase):
        """One alternation phase + its publish. 'users' begins the sweep
        and solves user rows against old item factors; 'items' solves item
        rows against the (globally published) updated users. Tests drive
        these directly to phase-lock multiple partitions in one process."""
        if self.lb is None:
            raise RuntimeError("half_step needs the bipartite boundary: "
                               "part.n_users is unset (load a graph with "
                               "-users / use the bipartite builders)")
        assert phase in ("users", "items")
        p = self.part
        if phase == "users":
            self._begin_sweep()
        if self.old_bf is not None:
            self.old_bf.copy_(self.old)  # RNE cast, one fused torch kernel
        if phase == "users":
            ng.cf_als_iter(_stream(), self.n0u, self.bin0s, self.n1u,
                           self.bin1s, self.n2u, self.bin2s, self.nbigu,
                           self.bin2vs, self.hubidx, self.gram, self.rhs_h,
                           p.row_ptr, p.col, p.weight, self.old,
                           self.new_part, p.row_left, self.K,
                           oldv_bf=self.old_bf)
        else:
            ng.cf_als_iter(_stream(), p.n0 - self.n0u,
                           self.bin0s[self.n0u:], p.n1 - self.n1u,
                           self.bin1s[self.n1u:], p.n2 - self.n2u,
                           self.bin2s[self.n2u * 2:], p.nbig - self.nbigu,
                           self.bin2vs[self.nbigu:], self.hubidx,
                           None if self.gram is None
                           else self.gram[self.nbigu * 64 * 64:],
                           None if self.rhs_h is None
                           else self.rhs_h[self.nbigu * 64:],
                           p.row_ptr, p.col, p.weight, self.old,
                           self.new_part, p.row_left, self.K,
                           oldv_bf=self.old_bf)
        self._publish()

    def step(self):
        p = self.part
        if self.lb is None:  # simultaneous Jacobi (no bipartite boundary)
            self._begin_sweep()
            if self.old_bf is not None:
                self.old_bf.copy_(self.old)
            ng.cf_als_iter(_stream(), p.n0, p.bin0, p.n1, p.bin1, p.n2,
                           p.bin2, p.nbig, p.bin2v, self.hubidx, self.gram,
                           self.rhs_h, p.row_ptr, p.col, p.weight, self.old,
                           self.new_part, p.row_left, self.K,
                           oldv_bf=self.old_bf)
            self._publish()
            return
        self.half_step("users")
        self.half_step("items")

    def vectors(self):
        return self.old.view(self.part.nv, self.K)
