"""Graph data layer vs networkx oracles."""
import numpy as np
import networkx as nx
import pytest

from multihop_offload_amd import CaseGraph, JobInstance


def _nx_graph(g):
    return nx.from_numpy_array(g.adj.astype(float))


def test_conflict_graph_matches_line_graph(small_case):
    g = small_case
    G = _nx_graph(g)
    L = nx.line_graph(G)
    # map our canonical link ids to nx line-graph nodes
    by_pair = {tuple(sorted(e)): i for i, e in enumerate(map(tuple, g.edges))}
    assert len(L.nodes) == g.num_links
    for (a, b) in L.nodes:
        i = by_pair[tuple(sorted((a, b)))]
        nbrs = {by_pair[tuple(sorted(x))] for x in L.neighbors((a, b))}
        lo, hi = g.conf_indptr[i], g.conf_indptr[i + 1]
        assert set(g.conf_indices[lo:hi].tolist()) == nbrs
    # conflict degrees
    degs = np.array([L.degree(n) for n in L.nodes])
    assert np.sum(degs) == np.sum(g.cf_degs)


def test_extended_graph_counts(small_case):
    g = small_case
    ext = g.ext
    n_comp = int(np.sum(g.roles < 2))
    assert ext.num_edges_ext == g.num_links + n_comp
    # virtual-edge maps cover exactly the computing nodes, ascending
    assert np.array_equal(ext.comp_nodes, np.nonzero(g.roles < 2)[0])
    assert np.array_equal(ext.maps_on_el,
                          g.num_links + np.arange(n_comp))
    # a virtual edge conflicts with exactly the real links at its node
    for k, u in enumerate(ext.comp_nodes):
        e = g.num_links + k
        lo, hi = ext.ext_indptr[e], ext.ext_indptr[e + 1]
        got = set(ext.ext_indices[lo:hi].tolist())
        want = set(g.link_matrix[u][g.adj[u] != 0].tolist())
        assert got == want


def test_extended_line_graph_matches_networkx(small_case):
    """Full structural check of the extended line graph against the
    reference construction (offloading_v3.py:262-339 semantics)."""
    g = small_case
    G = _nx_graph(g)
    N = g.num_nodes
    Ge = G.copy()
    for idx in range(N):
        if g.roles[idx] < 2:
            Ge.add_edge(idx, N + idx)
    L = nx.line_graph(Ge)
    ext = g.ext

    def canon_id(pair):
        a, b = sorted(pair)
        if b >= N:
            return int(ext.node_vedge[a])
        return int(g.link_matrix[a, b])

    assert len(L.nodes) == ext.num_edges_ext
    for node in L.nodes:
        i = canon_id(node)
        nbrs = {canon_id(x) for x in L.neighbors(node)}
        lo, hi = ext.ext_indptr[i], ext.ext_indptr[i + 1]
        assert set(ext.ext_indices[lo:hi].tolist()) == nbrs


def test_features(small_case, jobs_for):
    g, jobs = small_case, jobs_for
    x = g.ext.features(jobs)
    E = g.num_links
    assert x.shape == (g.ext.num_edges_ext, 4)
    assert np.all(x[:E, 0] == 0) and np.all(x[E:, 0] == 1)
    assert np.allclose(x[:E, 1], g.link_rates)
    assert np.allclose(x[E:, 1], g.proc_bws[g.ext.comp_nodes])
    # job arrivals sit on the source nodes' virtual edges only
    assert np.all(x[:E, 2] == 0)
    per_node = np.zeros(g.num_nodes)
    np.add.at(per_node, jobs.sources, jobs.rates * jobs.ul)
    assert np.allclose(x[E:, 2], per_node[g.ext.comp_nodes])
    # server flags
    servers = set(g.servers)
    for k, u in enumerate(g.ext.comp_nodes):
        assert x[E + k, 3] == (1.0 if u in servers else 0.0)


def test_job_sampling_distribution():
    g = CaseGraph(30, seed=3, gtype="ba")
    rng = np.random.RandomState(0)
    mobiles = np.arange(5, 30)
    for _ in range(20):
        jobs = JobInstance.sample(mobiles, 0.15, rng)
        assert int(0.3 * len(mobiles)) <= jobs.num_jobs < len(mobiles)
        assert np.all(np.isin(jobs.sources, mobiles))
        assert len(np.unique(jobs.sources)) == jobs.num_jobs
        assert np.all((jobs.rates >= 0.1 * 0.15) & (jobs.rates <= 0.5 * 0.15))
        assert np.all(jobs.ul == 100) and np.all(jobs.dl == 1)


@pytest.mark.parametrize("gtype", ["ba", "er", "ws", "poisson"])
def test_graph_families(gtype):
    g = CaseGraph(30, seed=11, gtype=gtype, m=4 if gtype == "poisson" else 2)
    assert g.num_links > 0
    assert g.adj.sum() == 2 * g.num_links


def test_mat_roundtrip(tmp_path):
    from multihop_offload_amd.datagen import generate_case
    import scipy.io as sio
    rng = np.random.RandomState(5)
    case, ns = generate_case(20, 123, "ba", rng=rng)
    path = str(tmp_path / "case.mat")
    sio.savemat(path, case)
    g = CaseGraph.from_mat(path)
    assert g.num_nodes == 20 and g.seed == 123 and g.m == 2
    assert g.mat_link_rate.shape == (g.num_links,)
    roles = case["nodes_info"][:, 0]
    assert np.array_equal(g.roles, roles)
    assert np.allclose(g.proc_bws, case["nodes_info"][:, 1])
    assert len(g.servers) == int(np.sum(roles == 1))
    assert len(g.relays) == int(np.sum(roles == 2))


def test_mobility_random_walk_and_topology_update():
    rng = np.random.RandomState(0)
    g = CaseGraph(25, seed=2, gtype="poisson", m=6)
    g.links_init(50.0, rng=rng)
    old_edges = {tuple(e) for e in g.edges.tolist()}
    old_lm = g.link_matrix.copy()
    adj, pos = g.random_walk(ss=0.05, n=5, rng=rng)
    m = g.topology_update(adj, pos)
    assert g.adj.sum() == 2 * g.num_links
    assert len(m) == g.num_links
    for l, (u, v) in enumerate(g.edges):
        if (u, v) in old_edges:
            assert m[l] == old_lm[u, v]
        else:
            assert m[l] == -1


def test_cf_radius_conflict_augmentation():
    """cf_radius > 0.5 adds interference conflicts between links whose
    endpoints lie within cf_radius x median link distance — checked
    against a literal networkx transcription of the reference rule
    (offloading_v3.py:193-224)."""
    import networkx as nx
    from scipy.spatial import distance_matrix as dmat

    g = CaseGraph(30, seed=5, gtype="poisson", m=4, cf_radius=1.5)
    base = CaseGraph(30, seed=5, gtype="poisson", m=4)   # same topology
    assert np.array_equal(g.edges, base.edges)
    pos = np.asarray(g.pos)

    # literal transcription
    d = dmat(pos, pos)
    link_dist = d[g.edges[:, 0], g.edges[:, 1]]
    intf = 1.5 * np.nanmedian(link_dist)
    gi = nx.Graph()
    gi.add_nodes_from(range(g.num_links))
    # base conflicts: shared endpoint
    for a in range(g.num_links):
        for b in range(a + 1, g.num_links):
            if set(g.edges[a]) & set(g.edges[b]):
                gi.add_edge(a, b)
    for l, (u, v) in enumerate(g.edges):
        near = np.union1d(np.nonzero(d[u] < intf)[0],
                          np.nonzero(d[v] < intf)[0])
        for w in near:
            for x in np.nonzero(g.adj[w])[0]:
                ll = g.link_matrix[w, x]
                if ll >= 0 and ll != l:
                    gi.add_edge(l, int(ll))

    for l in range(g.num_links):
        want = sorted(gi.neighbors(l))
        lo, hi = g.conf_indptr[l], g.conf_indptr[l + 1]
        got = sorted(g.conf_indices[lo:hi])
        assert got == want, (l, got, want)
    # augmentation strictly grows the conflict degree somewhere
    assert g.cf_degs.sum() > base.cf_degs.sum()


def test_cf_radius_engine_matches_oracle():
    """A cf_radius-augmented case runs through the batched engine with the
    same delays/gradients as the oracle (the conflict CSR feeds the fixed
    point everywhere)."""
    import torch
    from multihop_offload_amd.agent import ACOAgent, AgentConfig
    from multihop_offload_amd.env import AdhocCloudEnv
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.harness.common import run_method
    from multihop_offload_amd.models.chebconv import ChebConvStack

    rng = np.random.RandomState(7)
    g = CaseGraph(30, seed=5, gtype="poisson", m=4, cf_radius=1.5)
    g.links_init(50.0, rng=rng)
    g.add_relay(0)
    for s in (2, 3, 4):
        g.add_server(s, 300.0)
    for v in range(5, 30):
        if g.roles[v] == 0:
            g.set_mobile_bw(v, 10.0)
    jobs = JobInstance.sample(g.mobile_nodes, 0.15, rng)

    agent = ACOAgent(AgentConfig(seed=2), 10)
    for layer in agent.model.layers:
        with torch.no_grad():
            layer.weight.mul_(0.01)
    with torch.no_grad():
        agent.model.layers[-1].bias.fill_(0.5)
    env = AdhocCloudEnv(g)
    env.set_jobs(jobs)
    d_oracle = run_method("GNN", agent, env, 0.0, np.random.RandomState(0))

    model2 = ChebConvStack(K=2, dtype=torch.float64, seed=2)
    with torch.no_grad():
        for layer in model2.layers:
            layer.weight.mul_(0.01)
        model2.layers[-1].bias.fill_(0.5)
    eng = EpisodeEngine([g], model2, device="cpu", dtype=torch.float64)
    for p in model2.parameters():
        p.grad = None
    res = eng.gnn_episode(eng.pack_jobs([jobs]), train=True)
    np.testing.assert_allclose(res.delay_emp[0, :jobs.num_jobs], d_oracle,
                               rtol=1e-9)
    for p, go in zip(model2.parameters(), agent.memory[-1][0]):
        assert torch.allclose(p.grad, go, atol=1e-9)


def test_all_committed_mat_cases_load():
    """Every committed demo .mat case loads through the reference schema
    with consistent roles and a connected topology (C10 dataset check)."""
    import glob
    from scipy.sparse import csr_matrix
    from scipy.sparse.csgraph import connected_components

    mats = sorted(glob.glob("data_samples/**/*.mat", recursive=True))
    if not mats:
        import pytest
        pytest.skip("no committed .mat samples")
    assert len(mats) >= 10
    for path in mats:
        g = CaseGraph.from_mat(path)
        assert g.num_links == len(g.mat_link_rate)
        assert len(g.servers) >= 1 and len(g.mobile_nodes) >= 1
        ncomp, _ = connected_components(
            csr_matrix(g.adj.astype(float)), directed=False)
        assert ncomp == 1, path
        g.links_init(g.mat_link_rate, rng=np.random.RandomState(0))
        assert (g.link_rates >= 0).all()
        ext = g.ext
        assert ext.num_edges_ext == g.num_links + len(ext.comp_nodes)
