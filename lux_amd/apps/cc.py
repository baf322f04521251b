"""Connected Components app driver (reference CLI parity:
components/components.cc). Default engine: union-find (cc_engine.py —
one edge pass, identical max-label output); `-labelprop` selects the
reference-parity iterated label propagation (push model)."""
import sys

from .. import dist as dx
from ..cc_engine import CCUnionFindEngine
from ..engine import GraphPart
from ..push_engine import PushEngine
from .common import (ElapsedTimer, load_part, parse_input_args,
                     print_memory_estimate)


class CCBench:
    def __init__(self, part, labelprop=False):
        self.labelprop = labelprop
        if labelprop:
            self.eng = PushEngine(part, PushEngine.MODE_MAX)
        else:
            self.eng = CCUnionFindEngine(part)

    def step(self):
        if self.labelprop:
            self.eng.reset()
        self.eng.run()


def build_cc_bench(args, device):
    from ..engine import DeviceCSC
    # Twitter-2010-shaped synthetic, symmetrized:
    # connected components is defined on undirected graphs, so the bench
    # graph stores both directions of ne/2 generated pairs (ne total edges)
    nv = getattr(args, "nv", None) or 41652230
    if dx.world_size() > 1:  # rank-sliced build: graph/P per rank
        part = GraphPart.rmat_folded_sliced(nv, args.edges, dx.world_size(),
                                            dx.rank(), seed=args.seed,
                                            device=device, sym=True)
    else:
        full = DeviceCSC.rmat_folded(nv, args.edges, seed=args.seed,
                                     device=device, sym=True)
        part = GraphPart(full, 1, 0)
    return CCBench(part, labelprop=getattr(args, "labelprop", False)), part


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
    eng = PushEngine(part, PushEngine.MODE_MAX) if a.labelprop \
        else CCUnionFindEngine(part)
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
