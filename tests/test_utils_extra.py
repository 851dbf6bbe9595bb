"""Plotting, figures, MWIS, forward_gcn — auxiliary component coverage."""
import os

import numpy as np
import pandas as pd
import pytest
import torch


def test_mwis_is_independent_and_greedy():
    import networkx as nx
    from multihop_offload_amd.utils.mwis import local_greedy_search
    rng = np.random.RandomState(0)
    for seed in range(5):
        g = nx.gnp_random_graph(20, 0.25, seed=seed)
        adj = nx.to_numpy_array(g)
        w = rng.uniform(0.5, 2.0, 20)
        mwis, total = local_greedy_search(adj, w)
        # independence
        for u in mwis:
            for v in mwis:
                if u != v:
                    assert adj[u, v] == 0
        # maximality: every non-member has a member neighbour
        for v in range(20):
            if v not in mwis:
                assert any(adj[v, u] for u in mwis)
        assert np.isclose(total, w[list(mwis)].sum())


def test_figures_from_synthetic_csv(tmp_path):
    from multihop_offload_amd.harness import figures
    rng = np.random.RandomState(1)
    rows = []
    for n in (20, 30):
        for ni in range(4):
            for m in ("baseline", "local", "GNN"):
                rows.append({
                    "filename": "x.mat", "seed": 1, "num_nodes": n, "m": 2,
                    "num_mobile": n - 4, "num_servers": 3, "num_relays": 1,
                    "num_jobs": 5, "n_instance": ni, "Algo": m,
                    "runtime": 0.01, "tau": rng.uniform(10, 100),
                    "congest_jobs": rng.randint(0, 2),
                    "gnn_bl_ratio": rng.uniform(0.5, 2.0),
                    "gap_2_bl": rng.uniform(-10, 10)})
    csv = tmp_path / "Adhoc_test_data_x_load_0.15_T_1000.csv"
    pd.DataFrame(rows).to_csv(csv, index=False)
    figures.main(["--csv", str(csv), "--fig_dir", str(tmp_path / "fig")])
    pdfs = list((tmp_path / "fig").glob("*.pdf"))
    assert len(pdfs) == 3


def test_training_monitor_aggregation():
    from multihop_offload_amd.harness.figures import training_monitor
    df = pd.DataFrame({
        "fid": [0, 0, 1, 1], "method": ["GNN", "local", "GNN", "local"],
        "tau": [10.0, 20.0, 30.0, 40.0]})
    agg = training_monitor(df)
    assert agg.loc[0, "GNN"] == 10.0 and agg.loc[1, "local"] == 40.0


def test_plot_routes(tmp_path, small_case, jobs_for):
    from multihop_offload_amd.env import AdhocCloudEnv, apsp
    from multihop_offload_amd.utils.plotting import plot_routes
    g, jobs = small_case, jobs_for
    env = AdhocCloudEnv(g)
    env.set_jobs(jobs)
    _, dlist, dproc = env.dmtx_baseline()
    sp = apsp(g, dlist)
    np.fill_diagonal(sp, np.where(dproc > 0, dproc, g.T))
    env.offloading(sp, g.sp_hop)
    ldel, sdel, _ = env.run()
    path = plot_routes(g, env, ldel, sdel, 0, fig_dir=str(tmp_path))
    assert os.path.isfile(path)


def test_forward_gcn_variant(small_case, jobs_for):
    from multihop_offload_amd import ACOAgent
    from multihop_offload_amd.agent import AgentConfig
    from multihop_offload_amd.models.chebconv import ChebConvStack
    agent = ACOAgent(AgentConfig(seed=0), 10)
    agent.model = ChebConvStack(in_dim=3, K=2, dtype=torch.float64, seed=0)
    dm = agent.forward_gcn(small_case, jobs_for)
    N = small_case.num_nodes
    assert dm.shape == (N, N) and torch.isfinite(dm[0, 1])
    # wrong input width detected
    agent2 = ACOAgent(AgentConfig(seed=0), 10)
    with pytest.raises(ValueError):
        agent2.forward_gcn(small_case, jobs_for)


def test_dispatch_fails_loudly_without_extension(monkeypatch):
    """GPU execution must never silently fall back to eager torch when the
    native extension is missing."""
    from multihop_offload_amd.ops import dispatch
    monkeypatch.setattr(dispatch, "_HIP_EXT", None)
    monkeypatch.setattr(dispatch, "_HIP_TRIED", True)
    with pytest.raises(RuntimeError, match="HIP extension not built"):
        dispatch.require_hip()


def test_force_torch_escape_hatch(monkeypatch):
    """MHO_FORCE_TORCH=1 routes floyd_warshall to the torch reference even
    for CUDA tensors (debugging aid) — CPU tensors always use torch."""
    import torch as _t
    from multihop_offload_amd.ops import dispatch
    monkeypatch.setenv("MHO_FORCE_TORCH", "1")
    w = _t.full((1, 4, 4), float("inf"), dtype=_t.float64)
    for i in range(4):
        w[0, i, i] = 0
        w[0, i, (i + 1) % 4] = w[0, (i + 1) % 4, i] = 1.0
    d = dispatch.floyd_warshall(w)
    assert d[0, 0, 2] == 2.0


def test_fused_adam_state_and_lr_shim_cpu():
    """FusedAdam bookkeeping that needs no kernel: flat views, zero_grad,
    state_dict roundtrip, and the param_groups lr shim the trainer's lr
    schedules mutate."""
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.ops.functions import FusedAdam

    m = ChebConvStack(K=2, dtype=torch.float32, seed=1)
    opt = FusedAdam(m, lr=1e-4)
    # params/grads are views of the flat buffers
    p0 = next(m.parameters())
    assert p0.data.data_ptr() >= opt.flat_p.data_ptr()
    p0.grad.fill_(3.0)
    assert float(opt.flat_g.abs().sum()) > 0
    opt.zero_grad()
    assert float(opt.flat_g.abs().sum()) == 0.0
    # lr shim
    g = opt.param_groups[0]
    assert g["lr"] == 1e-4
    g["lr"] /= 4.0
    assert opt.lr == 2.5e-5
    # state roundtrip
    opt.m.fill_(0.5); opt.v.fill_(0.25); opt.step_dev.fill_(7)
    s = opt.state_dict()
    opt.m.zero_(); opt.v.zero_(); opt.step_dev.zero_(); opt.lr = 1.0
    opt.load_state_dict(s)
    assert float(opt.m.mean()) == 0.5 and float(opt.v.mean()) == 0.25
    assert int(opt.step_dev) == 7 and opt.lr == 2.5e-5


def test_figures_from_committed_results(tmp_path):
    """The committed result CSVs (C11) still drive the three paper
    figures."""
    import glob
    from multihop_offload_amd.harness import figures
    csvs = glob.glob("out_samples/Adhoc_test_*.csv")
    if not csvs:
        pytest.skip("no committed test CSV")
    figures.main(["--csv", csvs[0], "--fig_dir", str(tmp_path)])
    assert len(list(tmp_path.glob("*.pdf"))) == 3
