"""ChebConv actor network (torch).

Clean-room equivalent of the reference's 5-layer Spektral ChebConv model
(``gnn_offloading_agent.py:81-123``):

  * per layer: ``out = sum_k T_k(A) X @ W_k + b`` with the Chebyshev
    recurrence ``T_0 = X``, ``T_1 = A X``, ``T_k = 2 A T_{k-1} - T_{k-2}``;
  * hidden activation leaky_relu, output ReLU; dropout before each layer;
  * the support A is the raw adjacency of the extended line graph (the
    reference passes it unnormalised — ``gnn_offloading_agent.py:218,226``);
    a rescaled-Laplacian mode is available behind ``support='laplacian'``;
  * the shipped reference model uses Spektral's default K=1 (a per-node MLP —
    see SURVEY.md §2.4 K1); here K is explicit, default 2 per BASELINE.json;
  * Keras ``max_norm(1.0)`` kernel/bias constraints are applied after every
    optimizer step via ``apply_constraints()``.

The SpMV ``A @ X`` runs through ``ConflictCSR.spmv`` (gather + index_add):
batched graphs are block-diagonal, so one flat call covers a whole batch.
On GPU the fused HIP ChebConv kernel (ops/) replaces this layer-by-layer
torch path at inference; training keeps autograd semantics.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from ..queueing import ConflictCSR


def glorot_uniform_(w: torch.Tensor, gen: Optional[torch.Generator] = None):
    """Keras glorot_uniform for a (K, F_in, F_out) kernel: receptive field K,
    fan_in = K*F_in, fan_out = K*F_out."""
    k = w.shape[0] if w.dim() == 3 else 1
    fan_in = w.shape[-2] * k
    fan_out = w.shape[-1] * k
    limit = math.sqrt(6.0 / (fan_in + fan_out))
    with torch.no_grad():
        w.uniform_(-limit, limit, generator=gen)


class ChebConvLayer(nn.Module):
    def __init__(self, in_dim: int, out_dim: int, K: int, dtype=torch.float64,
                 gen: Optional[torch.Generator] = None):
        super().__init__()
        self.K = K
        self.weight = nn.Parameter(torch.empty(K, in_dim, out_dim, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(out_dim, dtype=dtype))
        glorot_uniform_(self.weight, gen)

    def forward(self, x: torch.Tensor, support: ConflictCSR) -> torch.Tensor:
        out = x @ self.weight[0]
        if self.K > 1:
            t_prev, t_cur = x, support.spmv(x)
            out = out + t_cur @ self.weight[1]
            for k in range(2, self.K):
                t_next = 2.0 * support.spmv(t_cur) - t_prev
                out = out + t_next @ self.weight[k]
                t_prev, t_cur = t_cur, t_next
        return out + self.bias

    @torch.no_grad()
    def apply_constraints(self):
        """Keras max_norm(1.0): kernel — clip the L2 norm over axis 0 (the K
        axis) at each (f_in, f_out); bias — clip the whole-vector norm."""
        w = self.weight
        norms = torch.sqrt((w * w).sum(dim=0, keepdim=True))
        w.mul_(torch.clamp(norms, max=1.0) / torch.clamp(norms, min=1e-12))
        bn = self.bias.norm().clamp(min=1e-12)
        self.bias.mul_(torch.clamp(bn, max=1.0) / bn)   # branchless, no sync


class ChebConvStack(nn.Module):
    """num_layer ChebConv layers: hidden width 32 + leaky_relu, final width 1
    + relu (``gnn_offloading_agent.py:87-110``)."""

    def __init__(self, in_dim: int = 4, hidden: int = 32, out_dim: int = 1,
                 num_layer: int = 5, K: int = 2, dropout: float = 0.0,
                 dtype=torch.float64, seed: Optional[int] = None):
        super().__init__()
        gen = None
        if seed is not None:
            gen = torch.Generator()
            gen.manual_seed(seed)
        dims = [in_dim] + [hidden] * (num_layer - 1) + [out_dim]
        self.layers = nn.ModuleList(
            ChebConvLayer(dims[i], dims[i + 1], K, dtype, gen)
            for i in range(num_layer))
        self.dropout = dropout

    def forward(self, x: torch.Tensor, support: ConflictCSR) -> torch.Tensor:
        h = x
        n = len(self.layers)
        for i, layer in enumerate(self.layers):
            if self.dropout > 0 and self.training:
                h = torch.dropout(h, self.dropout, True)
            h = layer(h, support)
            h = torch.relu(h) if i == n - 1 else torch.nn.functional.leaky_relu(h, 0.2)
        return h

    @torch.no_grad()
    def apply_constraints(self):
        for layer in self.layers:
            layer.apply_constraints()
