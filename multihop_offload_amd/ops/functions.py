"""autograd.Function wrappers around the fused HIP kernels."""

from __future__ import annotations

import torch

from . import dispatch


def cheb_compat(model) -> bool:
    """The fused ChebConv kernels cover the reference architecture family:
    any K (K<=2 on the tuned flagship kernels, K>=3 on the generic-K
    recurrence kernels), all feature widths <=32, no dropout."""
    if getattr(model, "dropout", 0.0) > 0:
        return False
    for layer in model.layers:
        K, fi, fo = layer.weight.shape
        if fi > 32 or fo > 32:
            return False
    return True


def cheb_lds_fits(Ee: int, K: int = 2) -> bool:
    rows_pad = (Ee + 15) & ~15
    return (3 * rows_pad * 33 + K * 32 * 32 + 32) * 4 <= 160 * 1024


class ChebStackFn(torch.autograd.Function):
    """Fused L-layer ChebConv stack: x (B,Ē,4) → λ (B,Ē) with per-layer
    activations saved for the fused backward.  Dispatches between the
    LDS-resident kernel (ops/hip/chebconv.hip) and the row-tiled
    large-graph kernels (ops/hip/chebconv_large.hip) by LDS fit."""

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
        ctx.K = K
        ctx.large = not cheb_lds_fits(eng.Ee, K)
        if K > 2:
            # generic-K recurrence kernels (3 LDS row buffers); no
            # large-graph variant — refuse loudly rather than fall back
            if ctx.large:
                raise RuntimeError(
                    f"fused ChebConv with K={K} does not fit LDS at "
                    f"Ee={eng.Ee}; no GPU path for this configuration "
                    "(set MHO_ALLOW_TORCH_GPU=1 to accept eager torch)")
            lam, acts, t1s = ext.cheb_kn_fwd(x, Wp, bp, eng.k_ext_indptr,
                                             eng.k_ext_base, eng.k_ext_cols,
                                             eng.k_ext_max_nnz)
            ctx.t1s = t1s
        elif ctx.large:
            lam, acts = ext.cheb_large_fwd(x, Wp, bp, eng.k_ext_indptr,
                                           eng.k_ext_base, eng.k_ext_cols)
        else:
            lam, acts, t1s = ext.cheb_fwd(x, Wp, bp, eng.k_ext_indptr,
                                          eng.k_ext_base, eng.k_ext_cols,
                                          eng.k_ext_max_nnz)
            ctx.t1s = t1s
        ctx.save_for_backward(acts, Wp)
        ctx.eng = eng
        ctx.shapes = [tuple(p.shape) for p in params]
        return lam

    @staticmethod
    def backward(ctx, dlam: torch.Tensor):
        acts, Wp = ctx.saved_tensors
        eng = ctx.eng
        ext = dispatch.require_hip()
        if ctx.K > 2:
            dW, db = ext.cheb_kn_bwd(dlam.contiguous(), acts, ctx.t1s, Wp,
                                     eng.k_ext_indptr, eng.k_ext_base,
                                     eng.k_ext_cols, eng.k_ext_max_nnz)
        elif ctx.large:
            dW, db = ext.cheb_large_bwd(dlam.contiguous(), acts, Wp,
                                        eng.k_ext_indptr, eng.k_ext_base,
                                        eng.k_ext_cols)
        else:
            dW, db = ext.cheb_bwd(dlam.contiguous(), acts, ctx.t1s, Wp,
                                  eng.k_ext_indptr, eng.k_ext_base,
                                  eng.k_ext_cols, eng.k_ext_max_nnz)
        if getattr(eng, "per_sample_request", False):
            eng._per_sample_raw = (dW, db)   # per-graph partials, pre-sum
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
            eng.T_arr.contiguous(), eng.k_E_arr, eng.N, eng.fp_iters,
            eng.delay_clamp)
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
            eng.k_E_arr, eng.fp_iters, eng.delay_clamp)
        return dlam, None


class FusedAdam:
    """Flat-buffer Adam with the reference's update semantics, fused into one
    kernel: grad scale → per-tensor clipnorm(1.0) →
    Adam(eps 1e-7) → Keras max_norm(1.0) constraints.  The model's parameter
    and gradient tensors become views of two flat buffers; ``flat_g`` is the
    single RCCL all-reduce payload for data-parallel training."""

    def __init__(self, model, lr=1e-4, betas=(0.9, 0.999), eps=1e-7,
                 constraints=True):
        import torch as _t
        params = list(model.parameters())
        n = sum(p.numel() for p in params)
        dev = params[0].device
        assert params[0].dtype == _t.float32, "FusedAdam is fp32"
        self.flat_p = _t.zeros(n, dtype=_t.float32, device=dev)
        self.flat_g = _t.zeros(n, dtype=_t.float32, device=dev)
        self.m = _t.zeros_like(self.flat_p)
        self.v = _t.zeros_like(self.flat_p)
        self.step_dev = _t.zeros(1, dtype=_t.int32, device=dev)
        segs = []
        off = 0
        with _t.no_grad():
            for p in params:
                k = p.numel()
                self.flat_p[off:off + k] = p.detach().reshape(-1)
                p.data = self.flat_p[off:off + k].view(p.shape)
                p.grad = self.flat_g[off:off + k].view(p.shape)
                if p.dim() == 3:
                    K, fi, fo = p.shape
                    segs.append([off, k, K, fi, fo])
                else:
                    segs.append([off, k, 0, 0, 0])
                off += k
        self.seg = _t.tensor(segs, dtype=_t.int32, device=dev)
        self.lr, self.betas, self.eps = lr, betas, eps
        self.constraints = constraints

    def zero_grad(self):
        self.flat_g.zero_()

    def step(self, scale: float = 1.0):
        ext = dispatch.require_hip()
        ext.fused_adam(self.flat_p, self.flat_g, self.m, self.v, self.seg,
                       self.step_dev, scale, self.lr, self.betas[0],
                       self.betas[1], self.eps, self.constraints)

    # -- snapshot/restore (guard rollbacks, resume) ---------------------------
    def state_dict(self):
        """Optimizer state snapshot (tensors cloned; device-agnostic
        restore).  The parameter buffer is NOT included — parameters are
        snapshotted by the caller like with torch optimizers."""
        return {"m": self.m.detach().clone(),
                "v": self.v.detach().clone(),
                "step_dev": self.step_dev.detach().clone(),
                "lr": self.lr}

    def load_state_dict(self, state):
        with torch.no_grad():
            self.m.copy_(state["m"])
            self.v.copy_(state["v"])
            self.step_dev.copy_(state["step_dev"])
        self.lr = state["lr"]

    @property
    def param_groups(self):
        """torch-optimizer-shaped lr access (the trainer's lr schedules
        mutate ``group["lr"]``)."""
        return [_FusedAdamGroup(self)]


class _FusedAdamGroup(dict):
    def __init__(self, opt):
        super().__init__(lr=opt.lr)
        self._opt = opt

    def __setitem__(self, key, value):
        super().__setitem__(key, value)
        if key == "lr":
            self._opt.lr = value
