"""lux-mi355x: an MI355X-native distributed graph-processing engine with the
capabilities of LuxGraph/Lux (pull/push vertex programs: PageRank, SSSP,
Connected Components, Collaborative Filtering) — built from scratch for
8x MI355X single-node: HIP/CDNA4 kernels + torch.distributed (RCCL) over
xGMI, one process per GPU.
"""
import os
import sys

# torch MUST be imported (and its bundled ROCm runtime stack loaded) before
# any other native library in this process: loading liblux_cpu.so / numpy's
# BLAS before torch breaks ROCm device detection process-wide ("no
# ROCm-capable device is detected" from HIP while torch.cuda still works) —
# bisected twice on MI355X boxes (gpurun_scripts/bisect_imports.sh,
# bisect2.sh). Importing torch here guarantees the order for every user of
# the package.
import torch  # noqa: F401

# repo root importable (for build.py lazy builds)
_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)

from .graph import Graph, Partition  # noqa: E402,F401

__version__ = "0.1.0"
