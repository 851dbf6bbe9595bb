"""Device dispatch for the compute kernels.

CPU tensors → pure-torch reference implementations (``torch_ref``).
GPU tensors → the gfx950 HIP extension (``ops/hip``), which is REQUIRED on
GPU: a missing extension raises instead of silently falling back to eager
torch, so GPU runs always exercise the native kernels.
"""

from __future__ import annotations

import os

import torch

from . import torch_ref

_HIP_EXT = None
_HIP_TRIED = False


def _load_hip():
    global _HIP_EXT, _HIP_TRIED
    if _HIP_TRIED:
        return _HIP_EXT
    _HIP_TRIED = True
    try:
        from . import hip_loader
        _HIP_EXT = hip_loader.load()
    except Exception:
        _HIP_EXT = None
    return _HIP_EXT


def hip_available() -> bool:
    return _load_hip() is not None


def require_hip():
    ext = _load_hip()
    if ext is None:
        raise RuntimeError(
            "multihop_offload_amd HIP extension not built — run "
            "`python setup.py build_ext --inplace` (gfx950). GPU execution "
            "without the native kernels is disabled by design.")
    return ext


_require_hip = require_hip


def floyd_warshall(w: torch.Tensor, n_arr=None) -> torch.Tensor:
    """Batched min-plus APSP.  ``n_arr`` (optional (B,) int32): per-graph
    effective node count for inert-padded batches — the kernel relaxes
    only the leading n×n submatrix (pad rows are +inf already)."""
    if w.is_cuda and os.environ.get("MHO_FORCE_TORCH") != "1":
        ext = _require_hip()
        return ext.floyd_warshall(w.contiguous(), n_arr)
    return torch_ref.floyd_warshall(w)
