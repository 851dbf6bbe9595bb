"""autograd.Function wrappers around the fused HIP kernels."""

from __future__ import annotations

import torch

from . import dispatch


class ActorHeadFn(torch.autograd.Function):
    """λ_ext (B,Ē) → delay matrix (B,N,N), fused forward/backward on GPU.

    Forward: contention fixed point (history saved) + congestion-fallback
    delays + symmetric scatter + diagonal (inf at relays).
    Backward: grad_dist (B,N,N) → δλ_ext via the hand-derived reverse pass
    (matches torch autograd semantics; see ops/hip/queueing.hip)."""

    @staticmethod
    def forward(ctx, lam_ext: torch.Tensor, eng):
        ext = dispatch.require_hip()
        dm, mu_hist = ext.actor_head_fwd(
            lam_ext.contiguous(), eng.k_conf_indptr, eng.k_conf_base,
            eng.k_conf_cols, eng.link_rates.contiguous(),
            eng.bw_comp.contiguous(), eng.k_edges, eng.node_vedge,
            float(eng.T), eng.N, eng.fp_iters)
        ctx.save_for_backward(lam_ext, mu_hist)
        ctx.eng = eng
        return dm

    @staticmethod
    def backward(ctx, grad_dist: torch.Tensor):
        lam_ext, mu_hist = ctx.saved_tensors
        eng = ctx.eng
        ext = dispatch.require_hip()
        dlam = ext.actor_head_bwd(
            grad_dist.contiguous(), lam_ext.contiguous(), mu_hist,
            eng.k_conf_indptr, eng.k_conf_base, eng.k_conf_cols,
            eng.link_rates.contiguous(), eng.bw_comp.contiguous(),
            eng.k_edges, eng.node_vedge, float(eng.T), eng.fp_iters)
        return dlam, None
