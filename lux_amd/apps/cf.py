"""Collaborative filtering app driver (reference parity:
col_filter/colfilter.cc; SGD matrix-factorization sweeps, rank -k)."""
import sys

from .. import dist as dx
from ..cf_engine import CFALSEngine, CFEngine
from ..engine import GraphPart
from .common import (ElapsedTimer, load_part, parse_input_args,
                     print_memory_estimate)

# NetFlix-prize shape (reference README.md:86: 497,959 V / 200,961,014 E)
NETFLIX_USERS = 480189
NETFLIX_ITEMS = 17770


def build_cf_bench(args, device):
    from ..engine import DeviceCSC
    ne = args.edges if args.edges != (1 << 31) else 200961014
    if dx.world_size() > 1:  # rank-sliced build: graph/P per rank
        part = GraphPart.bipartite_sliced(NETFLIX_USERS, NETFLIX_ITEMS, ne,
                                          dx.world_size(), dx.rank(),
                                          seed=args.seed, device=device)
    else:
        full = DeviceCSC.bipartite(NETFLIX_USERS, NETFLIX_ITEMS, ne,
                                   seed=args.seed, device=device)
        part = GraphPart(full, 1, 0)
    cls = CFALSEngine if getattr(args, "als", False) else CFEngine
    return cls(part, K=64), part


def main(argv=None):
    a = parse_input_args(sys.argv[1:] if argv is None else argv)
    dx.init_process_group("cuda")
    import torch
    local = dx.env_local_rank()
    torch.cuda.set_device(local)
    device = f"cuda:{local}"
    part = load_part(a, device, weighted=True)
    if a.users and part.n_users is None:  # -file graphs don't carry the
        part.n_users = a.users            # boundary; -users supplies it
    if dx.rank() == 0:
        print_memory_estimate(part.nv, part.ne, dx.world_size(),
                              weighted=True, k=a.k)
    eng = (CFALSEngine if a.als else CFEngine)(part, K=a.k)
    with ElapsedTimer():
        for _ in range(a.num_iter):
            eng.step()
    return eng


if __name__ == "__main__":
    main()
