"""End-to-end training quality gate: a short CPU training run must beat the
untrained policy on fresh instances (guards the whole gradient pipeline)."""
import numpy as np
import torch

from multihop_offload_amd.engine import EpisodeEngine
from multihop_offload_amd.models.chebconv import ChebConvStack
from multihop_offload_amd.harness.train_batched import build_training_cases


def _eval_tau(engine, gen, rounds=3, load=0.15):
    taus = []
    for _ in range(rounds):
        jobs = engine.sample_jobs(load, gen)
        res = engine.gnn_episode(jobs, train=False)
        taus.append(float(torch.nanmean(res.tau)))
    return float(np.mean(taus))


def test_short_training_improves_tau():
    cases = build_training_cases(20, 16, 8, 1000, 321, workers=2)
    model = ChebConvStack(K=2, dtype=torch.float64, seed=3)
    with torch.no_grad():
        for layer in model.layers:
            layer.weight.mul_(0.01)
        model.layers[-1].bias.fill_(0.5)
    engine = EpisodeEngine(cases, model, device="cpu", dtype=torch.float64)
    opt = torch.optim.Adam(model.parameters(), lr=1e-4, eps=1e-7)
    gen = torch.Generator().manual_seed(0)

    tau_before = _eval_tau(engine, gen)
    for _ in range(35):
        jobs = engine.sample_jobs(0.15, gen)
        for p in model.parameters():
            p.grad = None
        engine.gnn_episode(jobs, explore=0.05, gen=gen, train=True)
        with torch.no_grad():
            for p in model.parameters():
                if p.grad is not None:
                    p.grad /= engine.B
                    n = p.grad.norm().clamp(min=1e-12)
                    p.grad *= torch.clamp(n, max=1.0) / n
        opt.step()
        model.apply_constraints()
    tau_after = _eval_tau(engine, gen)
    assert tau_after < tau_before * 0.9, (tau_before, tau_after)


def test_nonfinite_grad_step_skipped():
    """The torch optimizer path zeroes a tensor's gradient when its norm
    is non-finite (the fp32 pole can emit inf/NaN): parameters must not
    be poisoned (the fused GPU kernel mirrors this — skip, not NaN)."""
    import torch
    from multihop_offload_amd.models.chebconv import ChebConvStack
    m = ChebConvStack(K=2, dtype=torch.float64, seed=1)
    before = [p.detach().clone() for p in m.parameters()]
    opt = torch.optim.Adam(m.parameters(), lr=1e-2, eps=1e-7)
    for p in m.parameters():
        p.grad = torch.full_like(p, float("nan"))
    # the trainer's guard logic
    with torch.no_grad():
        for p in m.parameters():
            n = p.grad.norm().clamp(min=1e-12)
            if not torch.isfinite(n):
                p.grad.zero_()
                continue
            p.grad *= torch.clamp(n, max=1.0) / n
    opt.step()
    for p, b in zip(m.parameters(), before):
        assert torch.isfinite(p).all()
        assert torch.allclose(p, b)    # zero grad → Adam no-op step
