"""SSSP app driver (reference parity: sssp/sssp.cc; unweighted hop
distances, push model with adaptive frontiers)."""
import sys

from .. import dist as dx
from ..engine import GraphPart
from ..push_engine import PushEngine
from .common import (ElapsedTimer, load_part, parse_input_args,
                     print_memory_estimate)


class SSSPBench:
    """One bench step = one full traversal to convergence from -start
    (reset + run; GTEPS counts ne per traversal, Graph500-style)."""

    def __init__(self, part, source):
        self.eng = PushEngine(part, PushEngine.MODE_MIN, source=source)

    def step(self):
        self.eng.reset()
        self.eng.run()


def build_sssp_bench(args, device):
    from ..engine import DeviceCSC
    if dx.world_size() > 1:  # rank-sliced build: graph/P per rank
        part = GraphPart.rmat_sliced(args.scale, args.edges,
                                     dx.world_size(), dx.rank(),
                                     seed=args.seed, device=device)
    else:
        full = DeviceCSC.rmat(args.scale, args.edges, seed=args.seed,
                              device=device)
        part = GraphPart(full, 1, 0)
    return SSSPBench(part, 0), part


def main(argv=None):
    a = parse_input_args(sys.argv[1:] if argv is None else argv)
    dx.init_process_group("cuda")
    import torch
    local = dx.env_local_rank()
    torch.cuda.set_device(local)
    device = f"cuda:{local}"
    part = load_part(a, device)
    if dx.rank() == 0:
        print_memory_estimate(part.nv, part.ne, dx.world_size())
    eng = PushEngine(part, PushEngine.MODE_MIN, source=a.start)
    with ElapsedTimer():
        iters = eng.run()
    if dx.rank() == 0:
        print(f"[lux] converged in {iters} iterations")
    if a.verbose and dx.rank() == 0 and hasattr(eng, "stats"):
        from ..trace import IterTrace
        tr = IterTrace()
        for row in eng.stats:
            tr.record(**row)
        print(tr.to_csv(), end="")
    if a.check:
        mistakes = eng.check()
        tag = "PASS" if mistakes == 0 else "FAIL"
        if dx.rank() == 0:
            print(f"[{tag}] {mistakes} mistakes")
    return eng


if __name__ == "__main__":
    main()
