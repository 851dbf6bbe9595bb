"""GPU (MI355X) tests: HIP kernels vs torch references, end-to-end engine."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="no GPU")


@needs_gpu
def test_hip_ext_loads():
    from multihop_offload_amd.ops import dispatch
    assert dispatch.hip_available(), \
        "HIP extension must be built in-tree (setup.py build_ext --inplace)"


def _rand_w(B, N, deg=4, seed=0):
    rng = np.random.RandomState(seed)
    w = np.full((B, N, N), np.inf, dtype=np.float32)
    for b in range(B):
        for _ in range(deg * N):
            i, j = rng.randint(N), rng.randint(N)
            if i != j:
                v = rng.uniform(0.01, 2.0)
                w[b, i, j] = w[b, j, i] = min(w[b, i, j], v)
        # ensure connectivity via a ring
        for i in range(N):
            j = (i + 1) % N
            v = rng.uniform(0.01, 2.0)
            w[b, i, j] = w[b, j, i] = min(w[b, i, j], v)
        np.fill_diagonal(w[b], 0.0)
    return torch.tensor(w)


@needs_gpu
@pytest.mark.parametrize("N", [20, 100, 199])
def test_fw_lds_kernel_vs_ref(N):
    from multihop_offload_amd.ops import dispatch, torch_ref
    w = _rand_w(8, N)
    want = torch_ref.floyd_warshall(w.double())
    got = dispatch.floyd_warshall(w.cuda()).cpu()
    assert torch.allclose(got.double(), want, rtol=1e-5, atol=1e-5)


@needs_gpu
@pytest.mark.parametrize("N", [256, 513])
def test_fw_tiled_kernel_vs_ref(N):
    from multihop_offload_amd.ops import dispatch, torch_ref
    w = _rand_w(2, N, deg=3)
    want = torch_ref.floyd_warshall(w.double())
    got = dispatch.floyd_warshall(w.cuda()).cpu()
    assert torch.allclose(got.double(), want, rtol=1e-4, atol=1e-4)


def _cases(n=20, B=8):
    from multihop_offload_amd.graphs import CaseGraph
    rng = np.random.RandomState(0)
    cases = []
    for b in range(B):
        g = CaseGraph(n, t_max=1000, seed=b + 1, gtype="ba")
        g.links_init(50.0, rng=rng)
        g.add_relay(0)
        for s in (2, 3, 4):
            g.add_server(s, 300.0)
        for v in range(5, n):
            g.set_mobile_bw(v, 10.0)
        cases.append(g)
    return cases


@needs_gpu
def test_engine_gpu_matches_cpu():
    """Same fp32 model, same injected jobs: GPU and CPU engines must agree
    on metrics and gradients (tolerances cover reduction-order effects)."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from tests.test_engine import _jobbatch_from
    from multihop_offload_amd.graphs import JobInstance

    cases = _cases()
    model_c = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    model_g = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    with torch.no_grad():
        for pc, pg in zip(model_c.parameters(), model_g.parameters()):
            pc.mul_(0.01)
            pg.copy_(pc)
        model_c.layers[-1].bias.fill_(0.5)
        model_g.layers[-1].bias.fill_(0.5)

    eng_c = EpisodeEngine(cases, model_c, device="cpu", dtype=torch.float32)
    eng_g = EpisodeEngine(cases, model_g, device="cuda", dtype=torch.float32)
    insts = [JobInstance.sample(c.mobile_nodes, 0.15,
                                np.random.RandomState(10 + i))
             for i, c in enumerate(cases)]
    jb_c = _jobbatch_from(eng_c, insts)
    jb_g = _jobbatch_from(eng_g, insts)

    res_c = eng_c.gnn_episode(jb_c, train=True)
    res_g = eng_g.gnn_episode(jb_g, train=True)
    tau_c, tau_g = res_c.tau.numpy(), res_g.tau.cpu().numpy()
    assert np.allclose(tau_c, tau_g, rtol=1e-3)
    assert np.isclose(float(res_c.loss_fn), float(res_g.loss_fn), rtol=1e-3)
    for pc, pg in zip(model_c.parameters(), model_g.parameters()):
        a, b = pc.grad.numpy(), pg.grad.cpu().numpy()
        denom = max(np.abs(a).max(), 1e-6)
        assert np.abs(a - b).max() / denom < 5e-3


@needs_gpu
def test_engine_gpu_train_step_moves_params():
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    cases = _cases()
    model = ChebConvStack(K=2, dtype=torch.float32, seed=0)
    with torch.no_grad():
        for layer in model.layers:
            layer.weight.mul_(0.01)
        model.layers[-1].bias.fill_(0.5)
    engine = EpisodeEngine(cases, model, device="cuda", dtype=torch.float32)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, eps=1e-7)
    gen = torch.Generator(device="cuda")
    gen.manual_seed(0)
    before = [p.detach().clone() for p in model.parameters()]
    for _ in range(2):
        jobs = engine.sample_jobs(0.15, gen)
        for p in model.parameters():
            p.grad = None
        res = engine.gnn_episode(jobs, train=True, gen=gen)
        opt.step()
        model.apply_constraints()
    assert torch.isfinite(res.tau).all()
    assert any(not torch.equal(b, p.detach())
               for b, p in zip(before, model.parameters()))


@needs_gpu
def test_baseline_local_episodes_gpu():
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    cases = _cases()
    model = ChebConvStack(K=2, dtype=torch.float32, seed=0)
    engine = EpisodeEngine(cases, model, device="cuda", dtype=torch.float32)
    gen = torch.Generator(device="cuda")
    gen.manual_seed(1)
    jobs = engine.sample_jobs(0.15, gen)
    rb = engine.baseline_episode(jobs)
    rl = engine.local_episode(jobs)
    assert torch.isfinite(rb.tau).all() and torch.isfinite(rl.tau).all()
    assert (rb.num_jobs == rl.num_jobs).all()


@needs_gpu
def test_engine_large_graph_fallback_paths():
    """A graph too large for some fused kernels must still run on GPU via
    the per-stage torch fallbacks (and the tiled FW), matching CPU."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.graphs import CaseGraph, JobInstance
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from tests.test_engine import _jobbatch_from
    import numpy as np

    rng = np.random.RandomState(0)
    g = CaseGraph(220, t_max=1000, seed=3, gtype="er")
    g.links_init(50.0, rng=rng)
    g.add_relay(0)
    for s in (2, 3, 4, 5):
        g.add_server(s, 300.0)
    for v in range(6, 220):
        if g.roles[v] == 0:
            g.set_mobile_bw(v, 10.0)
    cases = [g, g.clone_with_rates(np.full(g.num_links, 50.0), rng)]

    model_c = ChebConvStack(K=2, dtype=torch.float32, seed=1)
    model_g = ChebConvStack(K=2, dtype=torch.float32, seed=1)
    with torch.no_grad():
        for pc, pg in zip(model_c.parameters(), model_g.parameters()):
            pc.mul_(0.01)
            pg.copy_(pc)
        model_c.layers[-1].bias.fill_(0.5)
        model_g.layers[-1].bias.fill_(0.5)
    eng_c = EpisodeEngine(cases, model_c, device="cpu", dtype=torch.float32)
    eng_g = EpisodeEngine(cases, model_g, device="cuda", dtype=torch.float32)
    insts = [JobInstance.sample(c.mobile_nodes, 0.15,
                                np.random.RandomState(5 + i))
             for i, c in enumerate(cases)]
    res_c = eng_c.gnn_episode(_jobbatch_from(eng_c, insts), train=True)
    res_g = eng_g.gnn_episode(_jobbatch_from(eng_g, insts), train=True)
    assert np.allclose(res_c.tau.numpy(), res_g.tau.cpu().numpy(), rtol=1e-3)
    for pc, pg in zip(model_c.parameters(), model_g.parameters()):
        a, b = pc.grad.numpy(), pg.grad.cpu().numpy()
        assert np.abs(a - b).max() / max(np.abs(a).max(), 1e-6) < 1e-2


@needs_gpu
def test_fused_adam_matches_torch_path():
    """FusedAdam (one kernel) vs torch Adam + manual clip + constraints."""
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.ops.functions import FusedAdam

    torch.manual_seed(0)
    m1 = ChebConvStack(K=2, dtype=torch.float32, seed=7).cuda()
    m2 = ChebConvStack(K=2, dtype=torch.float32, seed=7).cuda()
    opt1 = FusedAdam(m1, lr=1e-3)
    opt2 = torch.optim.Adam(m2.parameters(), lr=1e-3, eps=1e-7)
    for it in range(5):
        grads = [torch.randn_like(p) * (10.0 if it % 2 else 0.01)
                 for p in m2.parameters()]
        opt1.zero_grad()
        for p, g in zip(m1.parameters(), grads):
            p.grad.copy_(g)
        opt1.step(scale=1.0)
        for p, g in zip(m2.parameters(), grads):
            n = g.norm().clamp(min=1e-12)
            p.grad = g * (torch.clamp(n, max=1.0) / n)
        opt2.step()
        m2.apply_constraints()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        a, b = p1.detach().cpu().numpy(), p2.detach().cpu().numpy()
        assert np.abs(a - b).max() < 1e-5, np.abs(a - b).max()


@needs_gpu
def test_ragged_link_counts_gpu_matches_cpu():
    """Ragged-E batch (distinct ER topologies) on the fused-kernel GPU path
    must match the CPU torch path."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.graphs import JobInstance
    from tests.test_engine import _er_case, _jobbatch_from

    cases = [_er_case(2), _er_case(5), _er_case(7)]
    assert len({c.num_links for c in cases}) > 1
    mc = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    mg = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    with torch.no_grad():
        for pc, pg in zip(mc.parameters(), mg.parameters()):
            pc.mul_(0.01)
            pg.copy_(pc)
        mc.layers[-1].bias.fill_(0.5)
        mg.layers[-1].bias.fill_(0.5)
    ec = EpisodeEngine(cases, mc, device="cpu", dtype=torch.float32)
    eg = EpisodeEngine(cases, mg, device="cuda", dtype=torch.float32)
    insts = [JobInstance.sample(c.mobile_nodes, 0.3,
                                np.random.RandomState(9 + i))
             for i, c in enumerate(cases)]
    rc = ec.gnn_episode(_jobbatch_from(ec, insts), train=True)
    rg = eg.gnn_episode(_jobbatch_from(eg, insts), train=True)
    assert np.allclose(rc.tau.numpy(), rg.tau.cpu().numpy(), rtol=1e-3)
    for pc, pg in zip(mc.parameters(), mg.parameters()):
        a, b = pc.grad.numpy(), pg.grad.cpu().numpy()
        assert np.abs(a - b).max() / max(np.abs(a).max(), 1e-6) < 1e-2


@needs_gpu
def test_large_mode_queueing_kernels_match_cpu():
    """A graph big enough to force the global-scratch critic/actor-head
    kernel modes must still match the CPU torch path (incl. gradients)."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.graphs import CaseGraph, JobInstance
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from tests.test_engine import _jobbatch_from

    rng = np.random.RandomState(0)
    g = CaseGraph(400, t_max=1000, seed=11, gtype="er")
    g.links_init(50.0, rng=rng)
    g.add_relay(0)
    for s in range(2, 10):
        g.add_server(s, 300.0)
    for v in range(10, 400):
        if g.roles[v] == 0:
            g.set_mobile_bw(v, 10.0)
    cases = [g]

    mc = ChebConvStack(K=2, dtype=torch.float32, seed=1)
    mg = ChebConvStack(K=2, dtype=torch.float32, seed=1)
    with torch.no_grad():
        for pc, pg in zip(mc.parameters(), mg.parameters()):
            pc.mul_(0.01)
            pg.copy_(pc)
        mc.layers[-1].bias.fill_(0.5)
        mg.layers[-1].bias.fill_(0.5)
    ec = EpisodeEngine(cases, mc, device="cpu", dtype=torch.float32)
    eg = EpisodeEngine(cases, mg, device="cuda", dtype=torch.float32)
    # the point of this test: the GPU engine must be using the kernels
    assert eg.hip_critic_ok and eg.hip_actor_ok and eg.hip_walk_ok
    insts = [JobInstance.sample(g.mobile_nodes, 0.2,
                                np.random.RandomState(3))]
    rc = ec.gnn_episode(_jobbatch_from(ec, insts), train=True)
    rg = eg.gnn_episode(_jobbatch_from(eg, insts), train=True)
    assert np.allclose(rc.tau.numpy(), rg.tau.cpu().numpy(), rtol=2e-3)
    for pc, pg in zip(mc.parameters(), mg.parameters()):
        a, b = pc.grad.numpy(), pg.grad.cpu().numpy()
        assert np.abs(a - b).max() / max(np.abs(a).max(), 1e-6) < 2e-2


@needs_gpu
def test_reference_checkpoint_in_engine_gpu():
    """The reference's shipped trained model (K=1 TF bundle) runs through
    the fused GPU engine and reproduces its local-collapse behavior."""
    import os
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.utils.tfckpt import load_reference_weights
    prefix = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "artifacts", "reference_ckpt",
        "cp-0000.ckpt")
    if not os.path.isfile(prefix + ".index"):
        pytest.skip("reference checkpoint not present")
    model = ChebConvStack(K=1, dtype=torch.float32)
    load_reference_weights(model, prefix)
    cases = _cases(n=30, B=16)
    engine = EpisodeEngine(cases, model, device="cuda", dtype=torch.float32)
    gen = torch.Generator(device="cuda")
    gen.manual_seed(0)
    jobs = engine.sample_jobs(0.15, gen)
    res = engine.gnn_episode(jobs, train=False)
    rl = engine.local_episode(jobs)
    # their trained policy collapses to local computing on these loads
    assert torch.allclose(res.tau, rl.tau, rtol=1e-3)


@needs_gpu
def test_per_sample_gradients_match_oracle():
    """engine.gnn_episode(per_sample=True) must reproduce the reference's
    per-instance gradient sets (oracle agent, fp32)."""
    from multihop_offload_amd import ACOAgent, AdhocCloudEnv
    from multihop_offload_amd.agent import AgentConfig
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.graphs import JobInstance
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from tests.test_engine import _case, _jobbatch_from, _wake

    g1, g2 = _case(seed=31), _case(seed=33)
    agent = ACOAgent(AgentConfig(T=1000, seed=5, dtype="float32"), 10)
    _wake(agent.model)
    model_g = ChebConvStack(K=2, dtype=torch.float32, seed=5)
    with torch.no_grad():
        for pc, pg in zip(agent.model.parameters(), model_g.parameters()):
            pg.copy_(pc)
    engine = EpisodeEngine([g1, g2], model_g, device="cuda",
                           dtype=torch.float32)
    j1 = JobInstance.sample(g1.mobile_nodes, 0.15, np.random.RandomState(1))
    j2 = JobInstance.sample(g2.mobile_nodes, 0.15, np.random.RandomState(2))
    res = engine.gnn_episode(_jobbatch_from(engine, [j1, j2]), train=True,
                             per_sample=True)
    psg = engine.last_per_sample_grads
    assert len(psg) == 2
    for b, (g, j) in enumerate(((g1, j1), (g2, j2))):
        env = AdhocCloudEnv(g)
        env.set_jobs(j)
        agent.forward_backward(env, 0.0, np.random.RandomState(0))
        oracle = agent.memory[-1][0]
        for got, want in zip(psg[b], oracle):
            a = got.cpu().numpy()
            w = want.cpu().numpy()
            assert np.abs(a - w).max() / max(np.abs(w).max(), 1e-6) < 1e-2


@needs_gpu
def test_padded_mixed_batch_gpu():
    """pad_to invariance on the HIP path: a case padded with inert relay
    nodes yields the same delays and gradients as the original, and a
    mixed-size batch in one engine sums the per-case gradients."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.graphs import JobInstance
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from tests.test_engine import _case, _wake

    g20, g25 = _case(seed=3, n=20), _case(seed=9, n=25)
    rng = np.random.RandomState(0)
    j20 = JobInstance.sample(g20.mobile_nodes, 0.15, rng)
    j25 = JobInstance.sample(g25.mobile_nodes, 0.15, rng)

    def run(cases, instances):
        model = ChebConvStack(K=2, dtype=torch.float32, seed=11)
        _wake(model)
        eng = EpisodeEngine(cases, model, device="cuda",
                            dtype=torch.float32)
        assert eng.use_hip
        for p in model.parameters():
            p.grad = None
        res = eng.gnn_episode(eng.pack_jobs(instances), train=True)
        torch.cuda.synchronize()
        grads = [p.grad.clone() for p in model.parameters()]
        return res, grads

    r_orig, g_orig = run([g20], [j20])
    r_pad, g_pad = run([g20.pad_to(25)], [j20])
    torch.testing.assert_close(r_pad.delay_emp[0, :j20.num_jobs],
                               r_orig.delay_emp[0, :j20.num_jobs],
                               rtol=1e-5, atol=1e-4, equal_nan=True)
    for a, b in zip(g_pad, g_orig):
        torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-5)

    r25, g25g = run([g25], [j25])
    r_mix, g_mix = run([g20.pad_to(25), g25], [j20, j25])
    torch.testing.assert_close(r_mix.delay_emp[0, :j20.num_jobs],
                               r_orig.delay_emp[0, :j20.num_jobs],
                               rtol=1e-5, atol=1e-4, equal_nan=True)
    torch.testing.assert_close(r_mix.delay_emp[1, :j25.num_jobs],
                               r25.delay_emp[0, :j25.num_jobs],
                               rtol=1e-5, atol=1e-4, equal_nan=True)
    for m, a, b in zip(g_mix, g_orig, g25g):
        torch.testing.assert_close(m, a + b, rtol=1e-3, atol=1e-4)


@needs_gpu
def test_engine_runner_gpu_matches_cpu_oracle():
    """The --engine harness path on the HIP kernels agrees with the CPU
    fp64 oracle at explore=0 (fp32 tolerances)."""
    from multihop_offload_amd import ACOAgent
    from multihop_offload_amd.agent import AgentConfig
    from multihop_offload_amd.env import AdhocCloudEnv
    from multihop_offload_amd.harness import common
    from multihop_offload_amd.graphs import JobInstance
    from tests.test_engine import _case, _wake

    g = _case(seed=13, n=20)
    jobs = JobInstance.sample(g.mobile_nodes, 0.15, np.random.RandomState(3))

    a_cpu = ACOAgent(AgentConfig(seed=4, dtype="float64"), 10)
    _wake(a_cpu.model)
    env = AdhocCloudEnv(g)
    env.set_jobs(jobs)
    d_oracle = common.run_method("GNN", a_cpu, env, 0.0,
                                 np.random.RandomState(0))

    a_gpu = ACOAgent(AgentConfig(seed=4, device="cuda",
                                 dtype="float32"), 10)
    with torch.no_grad():
        for pc, pg in zip(a_cpu.model.parameters(),
                          a_gpu.model.parameters()):
            pg.copy_(pc.to(torch.float32))
    runner = common.EngineRunner(a_gpu, g, seed=4)
    assert runner.engine.use_hip
    d_engine = runner.run_method("GNN", jobs, 0.0)
    np.testing.assert_allclose(d_engine, d_oracle, rtol=1e-3, atol=1e-2)


@needs_gpu
def test_fused_adam_state_dict_roundtrip():
    """FusedAdam state snapshot/restore (the guard's rollback unit):
    restoring params+state after divergent extra steps reproduces the
    checkpoint trajectory exactly."""
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.ops.functions import FusedAdam

    def make():
        m = ChebConvStack(K=2, dtype=torch.float32, seed=8).to("cuda")
        return m, FusedAdam(m, lr=1e-3)

    torch.manual_seed(0)
    m1, o1 = make()
    grads = [torch.randn_like(o1.flat_g) for _ in range(4)]

    def apply(o, g):
        o.zero_grad()
        o.flat_g.add_(g)
        o.step(scale=0.5)

    apply(o1, grads[0])
    snap_p = o1.flat_p.detach().clone()
    snap_s = o1.state_dict()
    apply(o1, grads[1])          # divergent step
    apply(o1, grads[2])
    # rollback
    with torch.no_grad():
        o1.flat_p.copy_(snap_p)
    o1.load_state_dict(snap_s)
    o1.param_groups[0]["lr"] = o1.param_groups[0]["lr"]   # shim no-op
    apply(o1, grads[3])
    after_rollback = o1.flat_p.detach().clone()

    # fresh trajectory: step0 then step3 directly
    torch.manual_seed(0)
    m2, o2 = make()
    apply(o2, grads[0])
    apply(o2, grads[3])
    torch.cuda.synchronize()
    torch.testing.assert_close(after_rollback, o2.flat_p,
                               rtol=1e-6, atol=1e-7)


@needs_gpu
def test_cheb_k3_gpu_matches_cpu():
    """Generic-K fused ChebConv (K=3): GPU gradients vs the CPU torch
    recurrence (models/chebconv.py:54-63)."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.graphs import JobInstance
    from tests.test_engine import _jobbatch_from

    cases = _cases()
    model_c = ChebConvStack(K=3, dtype=torch.float32, seed=7)
    model_g = ChebConvStack(K=3, dtype=torch.float32, seed=7)
    with torch.no_grad():
        for pc, pg in zip(model_c.parameters(), model_g.parameters()):
            pc.mul_(0.01)
            pg.copy_(pc)
        model_c.layers[-1].bias.fill_(0.5)
        model_g.layers[-1].bias.fill_(0.5)
    eng_c = EpisodeEngine(cases, model_c, device="cpu", dtype=torch.float32)
    eng_g = EpisodeEngine(cases, model_g, device="cuda", dtype=torch.float32)
    insts = [JobInstance.sample(c.mobile_nodes, 0.15,
                                np.random.RandomState(40 + i))
             for i, c in enumerate(cases)]
    res_c = eng_c.gnn_episode(_jobbatch_from(eng_c, insts), train=True)
    res_g = eng_g.gnn_episode(_jobbatch_from(eng_g, insts), train=True)
    assert np.allclose(res_c.tau.numpy(), res_g.tau.cpu().numpy(), rtol=1e-3)
    for pc, pg in zip(model_c.parameters(), model_g.parameters()):
        a, b = pc.grad.numpy(), pg.grad.cpu().numpy()
        denom = max(np.abs(a).max(), 1e-6)
        assert np.abs(a - b).max() / denom < 5e-3, (a - b)


@needs_gpu
def test_cheb_k4_forward_matches_cpu():
    """K=4 exercises the in-place Chebyshev ping-pong over >1 recurrence
    step (forward only — the shipped configs are K<=3)."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.graphs import JobInstance
    from tests.test_engine import _jobbatch_from
    cases = _cases(B=4)
    mc = ChebConvStack(K=4, dtype=torch.float32, seed=9)
    mg = ChebConvStack(K=4, dtype=torch.float32, seed=9)
    with torch.no_grad():
        for pc, pg in zip(mc.parameters(), mg.parameters()):
            pc.mul_(0.01)
            pg.copy_(pc)
    eng_c = EpisodeEngine(cases, mc, device="cpu", dtype=torch.float32)
    eng_g = EpisodeEngine(cases, mg, device="cuda", dtype=torch.float32)
    insts = [JobInstance.sample(c.mobile_nodes, 0.15,
                                np.random.RandomState(60 + i))
             for i, c in enumerate(cases)]
    res_c = eng_c.gnn_episode(_jobbatch_from(eng_c, insts), train=False)
    res_g = eng_g.gnn_episode(_jobbatch_from(eng_g, insts), train=False)
    assert np.allclose(res_c.tau.numpy(), res_g.tau.cpu().numpy(), rtol=1e-3)


@needs_gpu
def test_in_kernel_explore_uniform():
    """explore=1.0 in the decide kernel → destination uniform over the
    valid servers + local (statistical, counter-RNG)."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    cases = _cases(B=8)
    model = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    eng = EpisodeEngine(cases, model, device="cuda", dtype=torch.float32)
    eng.set_rng_seed(999)
    gen = torch.Generator(device="cuda")
    gen.manual_seed(0)
    counts = {}
    total = 0
    for _ in range(40):
        jobs = eng.sample_jobs(0.15, gen)
        res = eng.gnn_episode(jobs, explore=1.0, gen=gen, train=False)
        dm, *_ = eng.actor_forward(jobs)
        sp = eng.apsp(dm)
        uds = torch.diagonal(dm.detach(), dim1=1, dim2=2)
        dst, _ = eng.offload_decide(jobs, sp, uds, explore=1.0, gen=gen)
        d = dst.cpu().numpy()
        s = jobs.sources.cpu().numpy()
        m = jobs.mask.cpu().numpy()
        for b in range(eng.B):
            for j in range(eng.Jmax):
                if not m[b, j]:
                    continue
                total += 1
                key = "local" if d[b, j] == s[b, j] else int(d[b, j])
                counts[key] = counts.get(key, 0) + 1
    # 3 servers + local = 4 choices, uniform → 25% each.  "local" also
    # absorbs the rare greedy-pick-source case; generous tolerance.
    assert abs(counts.get("local", 0) / total - 0.25) < 0.08
    for s_ in (2, 3, 4):
        assert abs(counts.get(s_, 0) / total - 0.25) < 0.08


@needs_gpu
def test_in_kernel_prob_mode_matches_torch_distribution():
    """prob (softmax) sampling inside the kernel reproduces the torch
    multinomial distribution over choices (statistical)."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    cases = _cases(B=8)
    model = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    with torch.no_grad():
        for p in model.parameters():
            p.mul_(0.01)
        model.layers[-1].bias.fill_(0.5)
    eng_g = EpisodeEngine(cases, model, device="cuda", dtype=torch.float32)
    model_c = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    with torch.no_grad():
        for pc, pg in zip(model_c.parameters(), model.parameters()):
            pc.copy_(pg)
    eng_c = EpisodeEngine(cases, model_c, device="cpu", dtype=torch.float32)
    eng_g.set_rng_seed(31337)
    gen_g = torch.Generator(device="cuda")
    gen_g.manual_seed(5)
    gen_c = torch.Generator()
    gen_c.manual_seed(5)

    def dist(eng, gen, dev):
        counts = {}
        total = 0
        for r in range(30):
            g2 = torch.Generator(device=dev)
            g2.manual_seed(1000 + r)
            jobs = eng.sample_jobs(0.15, g2)
            dm, *_ = eng.actor_forward(jobs)
            sp = eng.apsp(dm)
            uds = torch.diagonal(dm.detach(), dim1=1, dim2=2)
            dst, _ = eng.offload_decide(jobs, sp, uds, gen=gen, prob=True)
            d = dst.cpu().numpy()
            s = jobs.sources.cpu().numpy()
            m = jobs.mask.cpu().numpy()
            for b in range(eng.B):
                for j in range(eng.Jmax):
                    if not m[b, j]:
                        continue
                    total += 1
                    key = ("local" if d[b, j] == s[b, j]
                           else int(d[b, j]))
                    counts[key] = counts.get(key, 0) + 1
        return {k: v / total for k, v in counts.items()}

    pg = dist(eng_g, gen_g, "cuda")
    pc = dist(eng_c, gen_c, "cpu")
    keys = set(pg) | set(pc)
    for k in keys:
        assert abs(pg.get(k, 0.0) - pc.get(k, 0.0)) < 0.06, (k, pg, pc)


@needs_gpu
def test_delay_clamp_gpu_matches_cpu():
    """delay_clamp threads identically through the HIP actor-head/critic
    kernels and the CPU torch path (values AND gradients)."""
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.graphs import JobInstance
    from tests.test_engine import _jobbatch_from
    cases = _cases()
    mc = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    mg = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    with torch.no_grad():
        for pc, pg in zip(mc.parameters(), mg.parameters()):
            pc.mul_(0.01)
            pg.copy_(pc)
        mc.layers[-1].bias.fill_(0.5)
        mg.layers[-1].bias.fill_(0.5)
    eng_c = EpisodeEngine(cases, mc, device="cpu", dtype=torch.float32,
                          delay_clamp=3.0)
    eng_g = EpisodeEngine(cases, mg, device="cuda", dtype=torch.float32,
                          delay_clamp=3.0)
    insts = [JobInstance.sample(c.mobile_nodes, 0.6,
                                np.random.RandomState(70 + i))
             for i, c in enumerate(cases)]
    res_c = eng_c.gnn_episode(_jobbatch_from(eng_c, insts), train=True)
    res_g = eng_g.gnn_episode(_jobbatch_from(eng_g, insts), train=True)
    assert np.allclose(res_c.tau.numpy(), res_g.tau.cpu().numpy(), rtol=1e-3)
    for pc, pg in zip(mc.parameters(), mg.parameters()):
        a, b = pc.grad.numpy(), pg.grad.cpu().numpy()
        denom = max(np.abs(a).max(), 1e-6)
        assert np.abs(a - b).max() / denom < 5e-3


@needs_gpu
def test_unit_mtx_deterministic_heterogeneous_jobs():
    """Heterogeneous (ul+dl) jobs over shared congested links: the kernel's
    packed atomicMax write order must reproduce the oracle's last-job-wins
    unit matrix exactly (and be deterministic across repeats)."""
    from multihop_offload_amd.engine import EpisodeEngine, JobBatch
    from multihop_offload_amd.models.chebconv import ChebConvStack
    cases = _cases(B=2)
    mc = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    mg = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    eng_c = EpisodeEngine(cases, mc, device="cpu", dtype=torch.float32)
    eng_g = EpisodeEngine(cases, mg, device="cuda", dtype=torch.float32)

    def het_jobs(eng, dev):
        B, J = eng.B, eng.Jmax
        # all jobs from the same mobile, heterogeneous ul → different
        # fallback units on the SAME links under forced congestion
        src = torch.full((B, J), int(cases[0].mobile_nodes[0]),
                         dtype=torch.int64, device=dev)
        mask = torch.ones(B, J, dtype=torch.bool, device=dev)
        rates = torch.full((B, J), 5.0, device=dev)
        ul = (50.0 + 10.0 * torch.arange(
            J, device=dev, dtype=torch.float32)).expand(B, J).contiguous()
        dl = torch.ones(B, J, device=dev)
        return JobBatch(sources=src, mask=mask, rates=rates, ul=ul, dl=dl)

    jb_c = het_jobs(eng_c, "cpu")
    jb_g = het_jobs(eng_g, "cuda")
    dst_c = torch.full_like(jb_c.sources, int(cases[0].servers[0]))
    dst_g = dst_c.cuda()
    sp_c, _, _ = eng_c.actor_forward(jb_c)
    sp_c = eng_c.apsp(sp_c)
    sp_g = sp_c.cuda()

    _, _, de_c, um_c, wr_c = eng_c._episode_eval(jb_c, dst_c, sp_c)
    um_gs = []
    for _ in range(3):
        _, _, de_g, um_g, wr_g = eng_g._episode_eval(jb_g, dst_g, sp_g)
        um_gs.append(um_g.cpu())
    assert torch.equal(um_gs[0], um_gs[1]) and torch.equal(um_gs[1],
                                                           um_gs[2])
    assert torch.equal(wr_c.cpu(), wr_g.cpu())
    # last-job-wins parity with the oracle path
    assert np.allclose(um_c.numpy(), um_gs[0].numpy(), rtol=1e-5, atol=1e-6)
    assert np.allclose(de_c.numpy(), de_g.cpu().numpy(), rtol=1e-4)


@needs_gpu
def test_train_batched_capture_smoke(tmp_path):
    """hipGraph-captured trainer: 30 steps with explore decay on a device
    scalar; must finish, write history, and move parameters."""
    from multihop_offload_amd.harness import train_batched
    hist = train_batched.main([
        "--steps", "30", "--batch", "16", "--nodes", "20",
        "--distinct", "4", "--T", "500", "--workers", "0",
        "--log_every", "10", "--save_every", "1000",
        "--guard_every", "0", "--capture",
        "--model_root", str(tmp_path), "--training_set", "CAPT"])
    assert any("tau" in h for h in hist)
