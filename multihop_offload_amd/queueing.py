"""Differentiable queueing model: contention fixed point + M/M/1-style delays.

Reference math (clean-room reimplementation, torch):
  * fixed point — ``gnn_offloading_agent.py:240-244`` (TF) and
    ``offloading_v3.py:500-506`` (NumPy): 10 iterations of
        busy   = clip(lambda / mu, 0, 1)
        nb     = A_conflict @ busy
        mu     = rates / (1 + nb)
    with mu0 = rates / (cf_deg + 1).
  * delays + congestion fallback — ``gnn_offloading_agent.py:245-254``:
        link_delay = 1/(mu - lambda);  where lambda > mu:
        link_delay = T * lambda / (101 * mu)
        node_delay = 1/(bw - lambda);  where lambda > bw:
        node_delay = T * lambda / (100 * bw)
    The empirical evaluator (``offloading_v3.py:537-547``) uses different
    fallback denominators: T*lambda/((ul+dl)*mu) per-job for links and
    T*load/(ul*bw) for servers — see ``env.run``.

All functions are pure tensor ops (autograd-friendly, CPU or GPU, optionally
batched by concatenation — the conflict CSR of a batch is block-diagonal, so
the same gather/index_add works unchanged).
"""

from __future__ import annotations

import torch


class ConflictCSR:
    """Sparse conflict adjacency as (row_ids, col_ids) for gather+index_add
    SpMV — autograd-friendly on CPU and GPU, batch = concatenation."""

    def __init__(self, indptr, indices, device="cpu"):
        indptr = torch.as_tensor(indptr, dtype=torch.int64)
        self.col = torch.as_tensor(indices, dtype=torch.int64, device=device)
        counts = indptr[1:] - indptr[:-1]
        self.row = torch.repeat_interleave(
            torch.arange(len(counts), dtype=torch.int64), counts).to(device)
        self.n = len(counts)

    def to(self, device):
        out = ConflictCSR.__new__(ConflictCSR)
        out.col = self.col.to(device)
        out.row = self.row.to(device)
        out.n = self.n
        return out

    def spmv(self, x: torch.Tensor) -> torch.Tensor:
        """y = A @ x for the 0/1 conflict adjacency (x: (n,) or (n,k))."""
        y = torch.zeros_like(x)
        return y.index_add(0, self.row, x[self.col])


def fixed_point_mu(lam: torch.Tensor, rates: torch.Tensor, cf_degs: torch.Tensor,
                   conf: ConflictCSR, iters: int = 10) -> torch.Tensor:
    """Contention fixed point: effective link service rates mu given per-link
    arrival rates lam.  Differentiable (the loop is unrolled by autograd)."""
    mu = rates / (cf_degs + 1.0)
    for _ in range(iters):
        busy = torch.clamp(lam / mu, 0.0, 1.0)
        nb = conf.spmv(busy)
        mu = rates / (1.0 + nb)
    return mu


def delay_with_fallback(lam: torch.Tensor, mu: torch.Tensor, T,
                        denom: float, cap: float = 0.0) -> torch.Tensor:
    """delay = 1/(mu-lam), replaced by T*lam/(denom*mu) where lam > mu.
    ``T`` may be a scalar or a tensor broadcastable against ``lam``
    (per-graph horizons).

    Matches the TF semantics: the congested entries are *overwritten* by the
    fallback (tensor_scatter_nd_update), so no gradient flows through the
    1/(mu-lam) branch at those entries.

    ``cap`` > 0 clamps the 1/(mu-lam) branch at ``cap`` (torch.clamp
    semantics: zero gradient where clamped).  This is the pole mitigation
    for training stability — as lam→mu⁻ the delay and its gradient
    1/(mu-lam)² blow up, which is the diagnosed driver of late-training
    collapse (docs/TRAINING.md).  The congested fallback branch (already
    bounded) is never clamped; 0 disables (reference semantics)."""
    congested = (lam - mu) > 0
    safe = torch.where(congested, torch.ones_like(mu), mu - lam)
    normal = 1.0 / safe
    if cap > 0:
        normal = normal.clamp(max=cap)
    fallback = T * lam / (denom * mu)
    return torch.where(congested, fallback, normal)


def actor_delays(lam_link, lam_node, rates, cf_degs, conf, proc_bws_comp, T,
                 iters: int = 10):
    """The actor-side delay head (``gnn_offloading_agent.py:229-254``):
    GNN-predicted per-link / per-computing-node arrival intensities →
    (link_delay[E], node_delay[n_comp]).  Differentiable."""
    mu = fixed_point_mu(lam_link, rates, cf_degs, conf, iters)
    link_delay = delay_with_fallback(lam_link, mu, T, 101.0)
    node_delay = delay_with_fallback(lam_node, proc_bws_comp, T, 100.0)
    return link_delay, node_delay


def delay_matrix(link_delay, node_delay, edges, comp_nodes, num_nodes):
    """Scatter link delays into a symmetric N×N matrix; diagonal = node delays
    (+inf at relays) (``gnn_offloading_agent.py:256-274``).  Differentiable."""
    N = num_nodes
    dm = link_delay.new_zeros((N, N))
    e0, e1 = edges[:, 0], edges[:, 1]
    dm = dm.index_put((e0, e1), link_delay)
    dm = dm.index_put((e1, e0), link_delay)
    diag = link_delay.new_full((N,), float("inf"))
    diag = diag.index_put((comp_nodes,), node_delay)
    dm = dm - torch.diag(torch.diagonal(dm)) + torch.diag(diag)
    return dm
