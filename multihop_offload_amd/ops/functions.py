"""autograd.Function wrappers around the fused HIP kernels."""

from __future__ import annotations

import torch

from . import dispatch


def cheb_compat(model) -> bool:
    """The fused ChebConv kernel covers the reference architecture family:
    K<=2, all feature widths <=32, no dropout."""
    if getattr(model, "dropout", 0.0) > 0:
        return False
    for layer in model.layers:
        K, fi, fo = layer.weight.shape
        if K > 2 or fi > 32 or fo > 32:
            return False
    return True


class ChebStackFn(torch.autograd.Function):
    """Fused L-layer ChebConv stack (ops/hip/chebconv.hip): x (B,Ē,4) →
    λ (B,Ē) with per-layer activations saved for the fused backward."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, eng, *params):
        ext = dispatch.require_hip()
        L = len(params) // 2
        K = params[0].shape[0]
        Wp = x.new_zeros(L, K, 32, 32)
        bp = x.new_zeros(L, 32)
        for l in range(L):
            w, b = params[2 * l], params[2 * l + 1]
            Wp[l, :w.shape[0], :w.shape[1], :w.shape[2]] = w
            bp[l, :b.shape[0]] = b
        lam, acts = ext.cheb_fwd(x, Wp, bp, eng.k_ext_indptr,
                                 eng.k_ext_base, eng.k_ext_cols,
                                 eng.k_ext_max_nnz)
        ctx.save_for_backward(acts, Wp)
        ctx.eng = eng
        ctx.shapes = [tuple(p.shape) for p in params]
        return lam

    @staticmethod
    def backward(ctx, dlam: torch.Tensor):
        acts, Wp = ctx.saved_tensors
        eng = ctx.eng
        ext = dispatch.require_hip()
        dW, db = ext.cheb_bwd(dlam.contiguous(), acts, Wp,
                              eng.k_ext_indptr, eng.k_ext_base,
                              eng.k_ext_cols, eng.k_ext_max_nnz)
        dW = dW.sum(dim=0)            # (L,K,32,32) summed over graphs
        db = db.sum(dim=0)            # (L,32)
        grads = []
        for l in range(len(ctx.shapes) // 2):
            kw, fi, fo = ctx.shapes[2 * l]
            grads.append(dW[l, :kw, :fi, :fo].contiguous())
            grads.append(db[l, :ctx.shapes[2 * l + 1][0]].contiguous())
        return (None, None, *grads)


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
            eng.T_arr.contiguous(), eng.N, eng.fp_iters)
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
            eng.k_edges, eng.node_vedge, eng.T_arr.contiguous(),
            eng.fp_iters)
        return dlam, None
