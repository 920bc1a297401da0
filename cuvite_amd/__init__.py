"""cuvite_amd — MI355X-native distributed Louvain community detection.

A from-scratch CDNA4/ROCm framework with the capabilities of pnnl/cuVite:
multi-phase Louvain modularity maximization over graphs partitioned across
the GPUs of one MI355X node, with hand-written HIP kernels for the
local-moving phase and modularity reduction, RCCL-over-xGMI halo exchange,
on-device coarsening, coloring-ordered moves, early termination, and
Vite-compatible binary graph I/O.
"""

__version__ = "0.1.0"

from .graph import Graph, DistGraph, Partition
from .louvain import louvain, LouvainConfig, LouvainResult

__all__ = [
    "Graph",
    "DistGraph",
    "Partition",
    "louvain",
    "LouvainConfig",
    "LouvainResult",
    "__version__",
]
