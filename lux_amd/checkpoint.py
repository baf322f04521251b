"""Vertex-state checkpoint/resume.

The reference has no checkpointing; its only persistence is
the .lux graph format. We add a vertex-state dump in the same binary-header
style, so a long PageRank/CF run (or a converged SSSP/CC labelling) can be
saved and resumed/analysed:

    u32 magic 'LUXS' (0x5358554C) | u32 dtype (0=f32, 1=u32) | u32 K |
    u32 nv | u64 iteration | payload nv*K elems

Distributed use: each rank saves its slice via `save_state(..., row_left,
data)` into per-rank files, or rank 0 saves the replicated array.
"""
import struct

import numpy as np

MAGIC = 0x5358554C
_DTYPES = {0: np.float32, 1: np.uint32}
_CODES = {np.dtype(np.float32): 0, np.dtype(np.uint32): 1,
          np.dtype(np.int32): 1}


def save_state(path, data, iteration=0):
    """data: numpy array [nv] or [nv, K] (f32 or u32/i32)."""
    arr = np.ascontiguousarray(data)
    k = 1 if arr.ndim == 1 else arr.shape[1]
    nv = arr.shape[0]
    code = _CODES[arr.dtype]
    with open(path, "wb") as f:
        f.write(struct.pack("<IIIIQ", MAGIC, code, k, nv, iteration))
        f.write(arr.tobytes())


def load_state(path):
    """Returns (array, iteration). Shape [nv] when K==1 else [nv, K]."""
    with open(path, "rb") as f:
        magic, code, k, nv, iteration = struct.unpack("<IIIIQ", f.read(24))
        if magic != MAGIC:
            raise IOError(f"{path}: not a lux state file")
        arr = np.frombuffer(f.read(nv * k * 4), _DTYPES[code]).copy()
    if k > 1:
        arr = arr.reshape(nv, k)
    return arr, iteration


def _labels_tensor(engine):
    lab = engine.labels
    return lab() if callable(lab) else lab  # LabelPullEngine method / tensor


def save_engine(path, engine, iteration=None):
    """Dump an engine's replicated state (rank 0 only in distributed runs)."""
    from . import dist as dx
    if dx.rank() != 0:
        return
    if hasattr(engine, "ranks"):  # PagerankEngine
        save_state(path, engine.ranks().cpu().numpy(), iteration or 0)
    elif hasattr(engine, "labels"):  # PushEngine / LabelPullEngine
        save_state(path, _labels_tensor(engine).cpu().numpy().view(np.uint32),
                   iteration if iteration is not None
                   else getattr(engine, "iterations", 0))
    elif hasattr(engine, "vectors"):  # CFEngine / CFALSEngine
        save_state(path, engine.vectors().cpu().numpy(), iteration or 0)
    else:
        raise TypeError(f"unknown engine {type(engine)}")


def resume_engine(path, engine):
    """Load saved state back into an engine's replicated array (all ranks).

    Restores vertex state (ranks / labels / latent vectors). For PushEngine
    the frontier is NOT part of the state (the reference's only persistence
    is the .lux graph); resume is for converged labellings and analysis —
    call reset() to restart a traversal."""
    import torch
    arr, iteration = load_state(path)
    if hasattr(engine, "ranks"):
        engine.old.copy_(torch.from_numpy(arr).to(engine.old.device))
        _refresh_slice(engine)
    elif hasattr(engine, "labels_t"):  # CCUnionFindEngine
        t = torch.from_numpy(arr.view(np.int32)).to(engine.labels_t.device)
        engine.labels_t.copy_(t)
        # a converged labelling (max vertex id per component) is itself a
        # valid union-find forest: every root r has label[r] == r
        engine.parent.copy_(t)
        engine.iterations = iteration
    elif hasattr(engine, "labels"):
        lab = _labels_tensor(engine)
        t = torch.from_numpy(arr.view(np.int32)).to(lab.device)
        p = engine.part
        if hasattr(engine, "labels_part"):  # PushEngine
            engine.labels.copy_(t)
            engine.labels_part.copy_(engine.labels.narrow(0, p.row_left,
                                                          p.vp))
            engine.iterations = iteration
            # restored labels can disagree with the BFS visited bitmap —
            # force a rebuild before the next push step
            engine._bits_stale = True
        else:  # LabelPullEngine
            engine.old.copy_(t)
            _refresh_slice(engine)
    elif hasattr(engine, "vectors"):
        t = torch.from_numpy(arr.reshape(-1)).to(engine.old.device)
        engine.old.copy_(t)
    return iteration


def _refresh_slice(engine):
    """Pipelined pull engines mirror their slice in cur_part and may hold an
    in-flight publish handle — resync after overwriting `old`."""
    h = getattr(engine, "_handle", None)
    if h is not None:
        if isinstance(h, list):  # per-peer publish handles
            for _q, w in h:
                w.wait()
        else:
            h.wait()
        engine._handle = None
    if hasattr(engine, "cur_part"):
        p = engine.part
        engine.cur_part.copy_(engine.old.narrow(0, p.row_left, p.vp))
