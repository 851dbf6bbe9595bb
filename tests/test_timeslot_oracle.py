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
