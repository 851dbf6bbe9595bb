"""Analytic evaluator vs the per-timeslot simulator (statistical oracle)."""
import numpy as np

from multihop_offload_amd import AdhocCloudEnv, JobInstance
from multihop_offload_amd.env import apsp, delay_empirical
from multihop_offload_amd.sim.timeslot import simulate


def test_analytic_delays_track_simulation(small_case):
    g = small_case
    rng = np.random.RandomState(7)
    jobs = JobInstance.sample(g.mobile_nodes, 0.10, rng)   # light load
    env = AdhocCloudEnv(g)
    env.set_jobs(jobs)
    _, dlist, dproc = env.dmtx_baseline()
    sp = apsp(g, dlist)
    np.fill_diagonal(sp, np.where(dproc > 0, dproc, g.T))
    env.offloading(sp, g.sp_hop)
    ldel, sdel, _ = env.run()
    analytic = delay_empirical(ldel, sdel)

    simulated, counts = simulate(g, jobs, env.flows, T=4000, seed=1,
                                 warmup=500)
    ok = counts > 20
    assert ok.sum() >= max(2, jobs.num_jobs // 2)
    ratio = simulated[ok] / analytic[ok]
    # same order of magnitude under stable load (M/D/1-vs-M/M/1-style gap)
    assert np.nanmedian(ratio) > 0.1 and np.nanmedian(ratio) < 10.0
    # rank correlation: jobs the analytic model calls slow should be slow
    if ok.sum() >= 4:
        a = analytic[ok]
        s = simulated[ok]
        ra = np.argsort(np.argsort(a)).astype(float)
        rs = np.argsort(np.argsort(s)).astype(float)
        corr = np.corrcoef(ra, rs)[0, 1]
        assert corr > 0.0


def test_trace_mode_and_plot_metrics(tmp_path, small_case, jobs_for):
    """trace=True conserves packets; facade plot_metrics saves the figure."""
    import os
    from multihop_offload_amd.env import AdhocCloudEnv, apsp
    from multihop_offload_amd.sim.timeslot import simulate

    g, jobs = small_case, jobs_for
    env = AdhocCloudEnv(g)
    env.set_jobs(jobs)
    _, dlist, dproc = env.dmtx_baseline()
    sp = apsp(g, dlist)
    np.fill_diagonal(sp, np.where(dproc > 0, dproc, g.T))
    env.offloading(sp, g.sp_hop)
    md, ct, tr = simulate(g, jobs, env.flows, T=400, seed=1, trace=True)
    assert tr["arrivals"].sum() - tr["departures"].sum() == \
        tr["pkts_in_network"][-1]
    assert tr["departures"].sum() >= ct.sum()

    from multihop_offload_amd import AdhocCloud
    net = AdhocCloud(20, t_max=400, seed=7, gtype="ba")
    rng = np.random.RandomState(42)
    net.links_init(50.0, rng=rng)
    net.add_relay(0); net.add_relay(1)
    for s in (2, 3, 4):
        net.add_server(s, 300.0)
    for n in range(5, 20):
        net.set_mobile_bw(n, 10.0)
    for s, r, ul, dl in zip(jobs.sources, jobs.rates, jobs.ul, jobs.dl):
        net.add_job(s, r, ul, dl)
    net.offloading(sp)
    arr, pkts, dep = net.plot_metrics(0, fig_dir=str(tmp_path))
    assert len(arr) == 400
    pngs = [f for f in os.listdir(tmp_path) if f.endswith(".png")]
    assert len(pngs) == 1 and "flow_packets" in pngs[0]
