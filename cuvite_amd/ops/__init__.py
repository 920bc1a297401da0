"""HIP kernel ops for gfx950 (MI355X).

The extension `_hip_ops` is built in-tree by setup.py / __graft_entry__.build()
with hipcc --offload-arch=gfx950. On a GPU box the HIP path is mandatory: if a
CUDA/HIP tensor reaches these wrappers without the extension present, we raise
instead of silently falling back to eager PyTorch.
"""

from __future__ import annotations

import os

import torch

_ext = None
_import_error: Exception | None = None
try:
    from . import _hip_ops as _ext  # type: ignore
except Exception as e:  # pragma: no cover - exercised only on broken builds
    _import_error = e


def available() -> bool:
    return _ext is not None


def _require():
    if _ext is None:
        raise RuntimeError(
            "cuvite_amd HIP extension (_hip_ops) is not built; run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_import_error!r}")
    return _ext


def local_move(inp):
    """HIP local-move iteration (see local_move.MoveInputs for semantics).
    Returns (target dense comm ids [nv], cluster_weight [nv])."""
    ext = _require()
    return ext.local_move(
        inp.rowptr, inp.tails, inp.weights, inp.curr_comm, inp.v_degree,
        inp.comm_size, inp.comm_degree, inp.comm_gid, float(inp.constant))


def modularity_parts(cluster_weight: torch.Tensor,
                     local_comm_degree: torch.Tensor) -> torch.Tensor:
    """HIP reduction of (sum cluster_weight, sum degree^2) in fp64."""
    ext = _require()
    return ext.modularity_parts(cluster_weight, local_comm_degree)
