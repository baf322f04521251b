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
    sweeps. K <= 64 (the MFMA tile grid is 4x4 of 16x16)."""

    def __init__(self, part: GraphPart, K=64):
        assert part.weight is not None, "CF needs a weighted graph"
        assert K <= 64, "ALS MFMA path covers K <= 64"
        self.part = part
        self.K = K
        part.build_bins()
        device = part.device
        v0 = math.sqrt(1.0 / K)
        self.old = torch.full((part.nv * K,), v0, dtype=F32, device=device)
        self.new_part = torch.empty(part.vp * K, dtype=F32, device=device)
        self.verts_elems = [v * K for v in part.verts_all]
        self.left_elems = [l * K for l in part.row_left_all]
        if part.nbig:
            self.hubidx = torch.full((part.vp,), -1, dtype=torch.int32,
                                     device=device)
            hubs = part.bin2v[:part.nbig].long()
            self.hubidx[hubs] = torch.arange(part.nbig, dtype=torch.int32,
                                             device=device)
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

    def step(self):
        p = self.part
        # seed with old slice: vertices with no in-edges keep their vector
        self.new_part.copy_(
            self.old.narrow(0, p.row_left * self.K, p.vp * self.K))
        if p.nbig:
            self.gram.zero_()
            self.rhs_h.zero_()
        if self.old_bf is not None:
            self.old_bf.copy_(self.old)  # RNE cast, one fused torch kernel
        ng.cf_als_iter(_stream(), p.n0, p.bin0, p.n1, p.bin1, p.n2, p.bin2,
                       p.nbig, p.bin2v, self.hubidx, self.gram, self.rhs_h,
                       p.row_ptr, p.col, p.weight, self.old, self.new_part,
                       p.row_left, self.K, oldv_bf=self.old_bf)
        dx.all_gather_slices(self.old, self.new_part, self.verts_elems,
                             self.left_elems, my_index=p.p)

    def vectors(self):
        return self.old.view(self.part.nv, self.K)
