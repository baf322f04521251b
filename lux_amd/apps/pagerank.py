"""PageRank app driver (reference parity: pagerank/pagerank.cc)."""
import sys

from .. import dist as dx
from ..engine import GraphPart, PagerankEngine
from .common import (ElapsedTimer, load_part, parse_input_args,
                     print_memory_estimate)


def build_pagerank_bench(args, device):
    from ..engine import DeviceCSC
    if dx.world_size() > 1:  # rank-sliced build: graph/P per rank
        part = GraphPart.rmat_sliced(args.scale, args.edges,
                                     dx.world_size(), dx.rank(),
                                     seed=args.seed, device=device)
    else:
        full = DeviceCSC.rmat(args.scale, args.edges, seed=args.seed,
                              device=device)
        part = GraphPart(full, 1, 0)
    return PagerankEngine(part), part


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
    eng = PagerankEngine(part)
    with ElapsedTimer():
        if a.verbose and dx.rank() == 0 and dx.world_size() == 1:
            from .common import run_traced
            run_traced(eng.step, a.num_iter, ne=part.ne)
        else:
            for _ in range(a.num_iter):
                eng.step()
    if a.verbose and dx.rank() == 0:
        r = eng.ranks()[:5].cpu().tolist()
        print("[lux] first ranks (pr/out_degree):", r)
    return eng


if __name__ == "__main__":
    main()
