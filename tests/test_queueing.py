"""Queueing math vs literal NumPy transcriptions of the reference equations
(offloading_v3.py:498-506, gnn_offloading_agent.py:240-254) + gradient checks."""
import numpy as np
import torch

from multihop_offload_amd.queueing import (
    ConflictCSR, actor_delays, delay_matrix, delay_with_fallback,
    fixed_point_mu)


def _naive_fixed_point(lam, rates, cf_degs, adj_dense, iters=10):
    mu = rates / (cf_degs + 1.0)
    for _ in range(iters):
        busy = np.clip(lam / mu, 0, 1.0)
        nb = adj_dense @ busy
        mu = rates / (1.0 + nb)
    return mu


def _conf_dense(g):
    E = g.num_links
    a = np.zeros((E, E))
    rows = np.repeat(np.arange(E), np.diff(g.conf_indptr))
    a[rows, g.conf_indices] = 1.0
    return a


def test_fixed_point_matches_naive(small_case):
    g = small_case
    rng = np.random.RandomState(0)
    lam = rng.uniform(0, 60, g.num_links)
    conf = ConflictCSR(g.conf_indptr, g.conf_indices)
    mu = fixed_point_mu(torch.tensor(lam), torch.tensor(g.link_rates),
                        torch.tensor(g.cf_degs), conf)
    want = _naive_fixed_point(lam, g.link_rates, g.cf_degs, _conf_dense(g))
    assert np.allclose(mu.numpy(), want, rtol=1e-12)


def test_delay_fallback_branches():
    lam = torch.tensor([1.0, 5.0, 10.0], dtype=torch.float64)
    mu = torch.tensor([4.0, 5.0, 5.0], dtype=torch.float64)
    d = delay_with_fallback(lam, mu, T=1000, denom=101.0)
    assert np.isclose(d[0].item(), 1.0 / 3.0)
    # tie lam == mu is NOT congested (strict >) → 1/0 = inf like the reference
    assert np.isinf(d[1].item())
    assert np.isclose(d[2].item(), 1000.0 * 10.0 / (101.0 * 5.0))


def test_fixed_point_gradcheck(small_case):
    g = small_case
    conf = ConflictCSR(g.conf_indptr, g.conf_indices)
    rates = torch.tensor(g.link_rates)
    cf = torch.tensor(g.cf_degs)
    lam = torch.tensor(np.random.RandomState(1).uniform(1, 30, g.num_links),
                       requires_grad=True)

    def f(x):
        return fixed_point_mu(x, rates, cf, conf).sum()

    assert torch.autograd.gradcheck(f, (lam,), eps=1e-6, atol=1e-6)


def test_delay_matrix_scatter(small_case):
    g = small_case
    ext = g.ext
    E = g.num_links
    rng = np.random.RandomState(2)
    ld = torch.tensor(rng.uniform(0.1, 2, E))
    nd = torch.tensor(rng.uniform(0.1, 2, len(ext.comp_nodes)))
    dm = delay_matrix(ld, nd, torch.tensor(g.edges),
                      torch.tensor(ext.comp_nodes), g.num_nodes).numpy()
    for l, (u, v) in enumerate(g.edges):
        assert dm[u, v] == dm[v, u] == ld[l].item()
    for k, u in enumerate(ext.comp_nodes):
        assert dm[u, u] == nd[k].item()
    for r in np.nonzero(g.roles == 2)[0]:
        assert np.isinf(dm[r, r])
    # non-edges are zero
    off = dm.copy()
    off[g.edges[:, 0], g.edges[:, 1]] = 0
    off[g.edges[:, 1], g.edges[:, 0]] = 0
    np.fill_diagonal(off, 0)
    assert np.all(off == 0)


def test_actor_delays_pipeline_gradflow(small_case):
    g = small_case
    ext = g.ext
    conf = ConflictCSR(g.conf_indptr, g.conf_indices)
    E, C = g.num_links, len(ext.comp_nodes)
    rng = np.random.RandomState(3)
    lam_l = torch.tensor(rng.uniform(0, 20, E), requires_grad=True)
    lam_n = torch.tensor(rng.uniform(0, 5, C), requires_grad=True)
    ld, nd = actor_delays(lam_l, lam_n, torch.tensor(g.link_rates),
                          torch.tensor(g.cf_degs), conf,
                          torch.tensor(g.proc_bws[ext.comp_nodes]), 1000)
    (ld.sum() + nd.sum()).backward()
    assert torch.isfinite(lam_l.grad).all()
    assert torch.isfinite(lam_n.grad).all()


def test_fixed_point_batched_equals_per_graph(small_case):
    """Block-diagonal flat batch == per-graph runs."""
    import numpy as np
    from multihop_offload_amd.graphs import CaseGraph
    g1 = small_case
    rng = np.random.RandomState(9)
    g2 = CaseGraph(20, seed=9, gtype="ba")
    g2.links_init(40.0, rng=rng)
    E = g1.num_links
    assert g2.num_links == E
    lam = torch.tensor(rng.uniform(0, 40, 2 * E))
    # flat batch CSR
    rows = np.concatenate([
        np.repeat(np.arange(E), np.diff(g1.conf_indptr)),
        np.repeat(np.arange(E), np.diff(g2.conf_indptr)) + E])
    cols = np.concatenate([g1.conf_indices, g2.conf_indices + E])
    conf = ConflictCSR.__new__(ConflictCSR)
    conf.row = torch.tensor(rows)
    conf.col = torch.tensor(cols)
    conf.n = 2 * E
    rates = torch.tensor(np.concatenate([g1.link_rates, g2.link_rates]))
    cf = torch.tensor(np.concatenate([g1.cf_degs, g2.cf_degs]))
    mu_flat = fixed_point_mu(lam, rates, cf, conf)
    for i, g in enumerate((g1, g2)):
        c = ConflictCSR(g.conf_indptr, g.conf_indices)
        mu = fixed_point_mu(lam[i * E:(i + 1) * E],
                            torch.tensor(g.link_rates),
                            torch.tensor(g.cf_degs), c)
        assert torch.allclose(mu_flat[i * E:(i + 1) * E], mu)


def test_delay_fallback_tensor_T():
    lam = torch.tensor([10.0, 10.0], dtype=torch.float64)
    mu = torch.tensor([5.0, 5.0], dtype=torch.float64)
    T = torch.tensor([700.0, 1000.0], dtype=torch.float64)
    d = delay_with_fallback(lam, mu, T, 101.0)
    assert np.isclose(d[0].item(), 700.0 * 10 / (101 * 5))
    assert np.isclose(d[1].item(), 1000.0 * 10 / (101 * 5))


def test_fixed_point_properties_random():
    """Analytic invariants of the contention fixed point over random
    conflict graphs: 0 < mu <= rates/(1) bounds, mu <= mu0 start,
    and adding load never increases any mu (monotonicity)."""
    from multihop_offload_amd.queueing import ConflictCSR, fixed_point_mu

    for seed in range(6):
        rng = np.random.RandomState(200 + seed)
        E = rng.randint(4, 40)
        # random symmetric conflict lists
        adj = np.triu(rng.rand(E, E) < 0.3, 1)
        adj = adj | adj.T
        indptr = np.zeros(E + 1, dtype=np.int64)
        indices = []
        for l in range(E):
            nbs = np.nonzero(adj[l])[0]
            indices.extend(nbs.tolist())
            indptr[l + 1] = len(indices)
        conf = ConflictCSR(indptr, np.asarray(indices, dtype=np.int64),
                           device="cpu")
        rates = torch.as_tensor(rng.uniform(5.0, 100.0, E))
        cf_degs = torch.as_tensor(adj.sum(1).astype(np.float64))
        lam = torch.as_tensor(rng.uniform(0.0, 10.0, E))

        mu = fixed_point_mu(lam, rates, cf_degs, conf, 10)
        assert torch.all(mu > 0)
        assert torch.all(mu <= rates + 1e-12)
        # more load somewhere -> no mu increases anywhere
        lam2 = lam.clone()
        j = rng.randint(E)
        lam2[j] += 5.0
        mu2 = fixed_point_mu(lam2, rates, cf_degs, conf, 10)
        assert torch.all(mu2 <= mu + 1e-9), (mu2 - mu).max()


def test_delay_fallback_cap_values_and_grad():
    """delay_clamp: the 1/(mu-lam) branch is clamped at `cap` with
    torch.clamp gradient semantics (zero where clamped); the congested
    fallback branch is never clamped."""
    from multihop_offload_amd.queueing import delay_with_fallback
    lam = torch.tensor([1.0, 9.99, 12.0], dtype=torch.float64,
                       requires_grad=True)
    mu = torch.tensor([10.0, 10.0, 10.0], dtype=torch.float64)
    out = delay_with_fallback(lam, mu, 1000.0, 101.0, cap=5.0)
    # normal branch: 1/9 ≈ 0.111; near-pole branch: 1/0.01 = 100 → capped
    assert torch.allclose(out[0], torch.tensor(1 / 9.0, dtype=torch.float64))
    assert float(out[1]) == 5.0
    # congested: fallback T*lam/(101*mu), NOT capped
    assert torch.allclose(out[2],
                          torch.tensor(1000.0 * 12.0 / (101.0 * 10.0),
                                       dtype=torch.float64))
    out.sum().backward()
    g = lam.grad
    assert abs(float(g[0]) - 1.0 / 81.0) < 1e-12   # d(1/(mu-lam))/dlam
    assert float(g[1]) == 0.0                       # clamped → zero grad
    assert abs(float(g[2]) - 1000.0 / (101.0 * 10.0)) < 1e-9


def test_delay_fallback_cap_zero_is_reference():
    from multihop_offload_amd.queueing import delay_with_fallback
    lam = torch.rand(50, dtype=torch.float64) * 20
    mu = torch.rand(50, dtype=torch.float64) * 20 + 0.1
    a = delay_with_fallback(lam, mu, 1000.0, 101.0)
    b = delay_with_fallback(lam, mu, 1000.0, 101.0, cap=0.0)
    assert torch.equal(a, b)
