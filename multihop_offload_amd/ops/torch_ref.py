"""Reference (pure-torch) implementations of the HIP kernels.

These are the semantics oracles: the GPU kernels in ``ops/hip`` are tested
against them, and they serve the CPU path.
"""

from __future__ import annotations

import torch


def floyd_warshall(w: torch.Tensor) -> torch.Tensor:
    """Batched min-plus Floyd–Warshall.  ``w``: (B,N,N) with +inf for
    non-edges and 0 diagonal.  Returns shortest-path distances (B,N,N).
    Equivalent to Dijkstra APSP for non-negative weights."""
    d = w.clone()
    n = d.shape[-1]
    for k in range(n):
        d = torch.minimum(d, d[:, :, k:k + 1] + d[:, k:k + 1, :])
    return d
