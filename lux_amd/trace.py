"""Per-iteration tracing/observability.

Parity+ with the reference's hand-rolled timing (`-verbose` loadTime/
compTime/updateTime per partition, sssp_gpu.cu:516-518; ELAPSED TIME,
pagerank.cc:118): a lightweight tracer records per-iteration phase timings
(hipEvents via torch.cuda.Event), frontier sizes and representation
choices, and dumps CSV + a GTEPS summary. Vendor-level counters come from
rocprofv3 (profiles/ holds committed summaries).
"""
import csv
import io


class IterTrace:
    """Collects per-iteration rows; timing uses CUDA events when enabled
    (adds ~us-level overhead; off by default in benchmarks)."""

    def __init__(self, time_phases=False):
        self.rows = []
        self.time_phases = time_phases
        self._events = None

    def phase_events(self, n):
        import torch
        if not self.time_phases:
            return None
        return [torch.cuda.Event(enable_timing=True) for _ in range(n)]

    def record(self, **kw):
        self.rows.append(kw)

    def to_csv(self, path=None):
        if not self.rows:
            return ""
        keys = sorted({k for r in self.rows for k in r})
        buf = io.StringIO()
        w = csv.DictWriter(buf, fieldnames=keys)
        w.writeheader()
        for r in self.rows:
            w.writerow(r)
        text = buf.getvalue()
        if path:
            with open(path, "w") as f:
                f.write(text)
        return text

    def summary(self, ne=None):
        n = len(self.rows)
        out = {"iterations": n}
        if n and "ms" in self.rows[0]:
            total = sum(r["ms"] for r in self.rows)
            out["total_ms"] = round(total, 3)
            out["ms_per_iter"] = round(total / n, 3)
            if ne:
                out["gteps"] = round(ne * n / total / 1e6, 3)
        return out
