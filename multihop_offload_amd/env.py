"""Single-case environment: offloading decisions, greedy routing, analytic
queueing evaluation.

Clean-room reimplementation of the decision/evaluation half of ``AdhocCloud``
(``/root/reference/src/offloading_v3.py:341-550``) on top of ``CaseGraph``.
This is the *oracle* path (numpy, single graph) used by the compatible
harnesses and as the ground truth for the batched device engine
(``engine.py``) and the HIP kernels (``ops/``).

Semantics preserved exactly (verified by tests against hand-derived cases):
  * baseline distance matrix: 1/link_rate off-diagonal, 1/proc_bw diagonal
    (``offloading_v3.py:341-361``)
  * offloading cost: max(sp_ul*ul, hops) + max(sp_dl*dl, hops) +
    max(proc*ul, 1), explore / argmin / softmax-sample
    (``offloading_v3.py:388-439``; the softmax prob mode reproduces the
    reference's high-cost-preferring behavior behind ``prob=True``)
  * greedy next-hop routing: argmin over neighbors of sp[nb, dst], ascending
    node-id tie-break (``offloading_v3.py:441-453``)
  * empirical evaluation with per-job congestion fallbacks
    (``offloading_v3.py:455-550``)
"""

from __future__ import annotations

import dataclasses
from typing import List, Optional

import numpy as np

from .graphs import CaseGraph, JobInstance


@dataclasses.dataclass
class Flow:
    """Reference ``Flow`` (``offloading_v3.py:140-150``), route as node list."""
    src: int
    dst: int
    route: List[int]
    nhop: int


def softmax(x: np.ndarray) -> np.ndarray:
    """Reference ``util.softmax`` (``util.py:113-116``) — NOT max-subtracted,
    NOT negated: prefers high-cost entries. Kept verbatim for parity."""
    ex = np.exp(x)
    return ex / ex.sum()


def apsp(g: CaseGraph, link_weights: Optional[np.ndarray] = None) -> np.ndarray:
    """All-pairs shortest path distances over the connectivity graph.
    ``link_weights=None`` → hop counts. Reference: ``util.py:101-110``
    (networkx Dijkstra); here scipy's C Dijkstra on the canonical CSR."""
    from scipy.sparse import csr_matrix
    from scipy.sparse.csgraph import shortest_path
    if link_weights is None:
        return g.sp_hop
    N = g.num_nodes
    w = np.zeros((N, N))
    w[g.edges[:, 0], g.edges[:, 1]] = link_weights
    w[g.edges[:, 1], g.edges[:, 0]] = link_weights
    mask = g.adj != 0
    data = csr_matrix((w[mask], np.nonzero(mask)), shape=(N, N))
    return shortest_path(data, method="D", directed=False)


class AdhocCloudEnv:
    """Decision + evaluation wrapper around one ``CaseGraph`` and one
    ``JobInstance``."""

    def __init__(self, g: CaseGraph):
        self.g = g
        self.jobs: Optional[JobInstance] = None
        self.flows: List[Flow] = []

    # -- job management (offloading_v3.py:243-250) ----------------------------
    def set_jobs(self, jobs: JobInstance):
        self.jobs = jobs
        self.flows = []

    # -- baseline distance matrix (offloading_v3.py:341-361) ------------------
    def dmtx_baseline(self):
        g = self.g
        N = g.num_nodes
        dmtx = np.full((N, N), np.inf)
        with np.errstate(divide="ignore"):
            dproc = 1.0 / g.proc_bws            # inf at relays (bw=0)
            dlist = 1.0 / g.link_rates
        np.fill_diagonal(dmtx, dproc)
        dmtx[g.edges[:, 0], g.edges[:, 1]] = dlist
        dmtx[g.edges[:, 1], g.edges[:, 0]] = dlist
        return dmtx, dlist, dproc

    # -- local computing (offloading_v3.py:363-386) ---------------------------
    def local_compute(self, unit_delay_servers: np.ndarray):
        jobs = self.jobs
        self.flows = [Flow(int(s), int(s), [int(s), int(s)], 0)
                      for s in jobs.sources]
        delays = np.maximum(unit_delay_servers[jobs.sources] * jobs.ul, 1.0)
        return jobs.sources.copy(), delays

    # -- offloading decision (offloading_v3.py:388-439) -----------------------
    def offloading(self, spmtx_in: np.ndarray, hpmtx: np.ndarray,
                   explore: float = 0.0, prob: bool = False,
                   rng: Optional[np.random.RandomState] = None):
        rng = rng or np.random
        g, jobs = self.g, self.jobs
        servers = np.asarray(g.servers, dtype=np.int64)
        S = len(servers)
        unit_delay_servers = np.diagonal(spmtx_in)
        spmtx = spmtx_in.copy()
        np.fill_diagonal(spmtx, 0)

        src = jobs.sources
        local_delay = unit_delay_servers[src] * jobs.ul                    # (J,)
        ul_d = np.maximum(spmtx[src][:, servers] * jobs.ul[:, None],
                          hpmtx[src][:, servers])                          # (J,S)
        dl_d = np.maximum(spmtx[servers][:, src].T * jobs.dl[:, None],
                          hpmtx[servers][:, src].T)
        proc_d = np.maximum(unit_delay_servers[servers][None, :] * jobs.ul[:, None],
                            1.0)
        server_delays = ul_d + dl_d + proc_d                               # (J,S)
        costs = np.concatenate([server_delays, local_delay[:, None]], axis=1)

        decisions, delays = [], []
        self.flows = []
        for j in range(jobs.num_jobs):
            if rng.uniform(0, 1) < explore:
                jidx = rng.choice(S + 1)
            elif not prob:
                jidx = int(np.argmin(costs[j]))
            else:
                jidx = rng.choice(S + 1, p=softmax(costs[j]))
            s = int(src[j])
            if jidx < S:
                dst = int(servers[jidx])
                route, nhop = self.routing(s, dst, spmtx)
                job_delay = server_delays[j, jidx]
            else:
                dst, route, nhop = s, [s, s], 0
                job_delay = local_delay[j]
            self.flows.append(Flow(s, dst, route, nhop))
            decisions.append(dst)
            delays.append(job_delay)
        return np.asarray(decisions), np.asarray(delays)

    # -- greedy next-hop walk (offloading_v3.py:441-453) ----------------------
    def routing(self, src: int, dst: int, spmtx: np.ndarray):
        g = self.g
        route = [src]
        node, nhop = src, 0
        while node != dst:
            lo, hi = g.adj_indptr[node], g.adj_indptr[node + 1]
            nbs = g.adj_indices[lo:hi]
            node = int(nbs[np.argmin(spmtx[nbs, dst])])
            nhop += 1
            route.append(node)
            if nhop > g.num_nodes:
                raise RuntimeError("greedy route failed to terminate")
        return route, nhop

    # -- route → link-id sequences --------------------------------------------
    def route_links(self, flow: Flow) -> np.ndarray:
        """Link ids along a route (empty for local computing)."""
        g = self.g
        if flow.src == flow.dst:
            return np.empty(0, dtype=np.int64)
        r = np.asarray(flow.route)
        return g.link_matrix[r[:-1], r[1:]]

    # -- analytic queueing evaluation (offloading_v3.py:455-550) --------------
    def run(self):
        g, jobs = self.g, self.jobs
        N, E, J = g.num_nodes, g.num_links, jobs.num_jobs
        assert len(self.flows) == J
        server_delay = np.full((N, J), np.nan)
        link_delay = np.full((E, J), np.nan)
        link_load = np.zeros((E, J))
        server_load = np.zeros(N)

        ul_rate = jobs.ul * jobs.rates
        dl_rate = jobs.dl * jobs.rates
        for j, flow in enumerate(self.flows):
            links = self.route_links(flow)
            np.add.at(link_load, (links, np.full(len(links), j)),
                      ul_rate[j] + dl_rate[j])
            server_load[flow.dst] += ul_rate[j]

        link_lambda = link_load.sum(axis=1)
        link_mu = g.link_rates / (g.cf_degs + 1.0)
        for _ in range(10):
            busy = np.clip(link_lambda / link_mu, 0, 1.0)
            nb = np.zeros(E)
            np.add.at(nb, _conf_rows(g), busy[g.conf_indices])
            link_mu = g.link_rates / (1.0 + nb)

        unit_mtx = np.full((N, N), np.nan)
        for j, flow in enumerate(self.flows):
            job_tot = jobs.ul[j] + jobs.dl[j]
            nhop = float(flow.nhop)
            for lidx in self.route_links(flow):
                gap = link_mu[lidx] - link_lambda[lidx]
                if gap <= 0:
                    unit = float(g.T) * link_lambda[lidx] / (job_tot * link_mu[lidx])
                else:
                    unit = 1.0 / gap
                u, v = g.edges[lidx]
                unit_mtx[u, v] = unit_mtx[v, u] = unit
                link_delay[lidx, j] = (max(jobs.ul[j] * unit, nhop)
                                       + max(jobs.dl[j] * unit, nhop))
            dst = flow.dst
            gap = g.proc_bws[dst] - server_load[dst]
            if gap <= 0:
                unit = float(g.T) * server_load[dst] / (jobs.ul[j] * g.proc_bws[dst])
            else:
                unit = 1.0 / gap
            unit_mtx[dst, dst] = unit
            server_delay[dst, j] = max(jobs.ul[j] * unit, 1.0)
        return link_delay, server_delay, unit_mtx


def _conf_rows(g: CaseGraph) -> np.ndarray:
    # Cached ON the graph instance (a module-level id(g)-keyed dict is
    # unsafe: CPython reuses ids after garbage collection, so a long
    # harness run could silently pick up another topology's row indices).
    got = getattr(g, "_conf_rows_cache", None)
    if got is None or len(got) != len(g.conf_indices):
        got = np.repeat(np.arange(g.num_links), np.diff(g.conf_indptr))
        g._conf_rows_cache = got
    return got


def delay_empirical(link_delay: np.ndarray, server_delay: np.ndarray) -> np.ndarray:
    """Per-job total delay: nansum over links + nansum over servers
    (``AdHoc_train.py:140,153``)."""
    return np.nansum(link_delay, axis=0) + np.nansum(server_delay, axis=0)


class AdhocCloud(CaseGraph):
    """Drop-in facade with the reference's ``AdhocCloud`` API
    (``offloading_v3.py:29-550``): constructor accepting a graph family or
    a ``.mat`` path as ``gtype``, imperative job management, and the
    decision/evaluation methods delegating to :class:`AdhocCloudEnv`.

    New code should use :class:`~multihop_offload_amd.graphs.CaseGraph` +
    :class:`AdhocCloudEnv` (or the batched engine) directly; this class
    exists so reference-shaped scripts port without edits.
    """

    def __init__(self, num_nodes, t_max=1000, seed=3, m=2, pos=None,
                 cf_radius=0.0, gtype="ba", trace=False):
        if ".mat" in str(gtype):
            base = CaseGraph.from_mat(gtype, t_max=t_max, cf_radius=cf_radius)
            self.__dict__.update(base.__dict__)   # adopt graph + roles/bws
        else:
            super().__init__(num_nodes, t_max=t_max, seed=seed, m=m,
                             gtype=gtype, pos=pos, cf_radius=cf_radius)
        self.trace = trace
        self._env = AdhocCloudEnv(self)
        self._job_list = []

    # -- job management (offloading_v3.py:243-250) ----------------------------
    def add_job(self, src, rate=0.1, ul=100, dl=1):
        self._job_list.append((int(src), float(rate), float(ul), float(dl)))
        self._sync_jobs()

    def clear_all_jobs(self):
        self._job_list = []
        self._env.jobs = None
        self._env.flows = []

    @property
    def num_jobs(self):
        return len(self._job_list)

    def _sync_jobs(self):
        a = np.asarray(self._job_list, dtype=np.float64).reshape(-1, 4)
        self._env.set_jobs(JobInstance(
            sources=a[:, 0].astype(np.int64), rates=a[:, 1],
            ul=a[:, 2], dl=a[:, 3]))

    @property
    def flows(self):
        return self._env.flows

    # -- decision / evaluation delegates --------------------------------------
    def graph_expand(self):
        return self.ext

    def dmtx_baseline(self):
        return self._env.dmtx_baseline()

    def local_compute(self, unit_delay_servers):
        return self._env.local_compute(unit_delay_servers)

    def offloading(self, spmtx_in, hpmtx=None, explore=0.0, prob=False):
        if hpmtx is None:
            hpmtx = self.sp_hop
        return self._env.offloading(spmtx_in, hpmtx, explore, prob)

    def routing(self, flow, spmtx, *rest):
        """Reference form ``routing(flow, spmtx)``; also accepts
        ``routing(src, dst, spmtx)`` with plain node ids."""
        if rest:
            src, dst, spmtx = int(flow), int(spmtx), rest[0]
        else:
            src, dst = int(flow.src), int(flow.dst)
        return self._env.routing(src, dst, spmtx)

    def run(self):
        return self._env.run()

    def plot_routes(self, link_delays, node_delays, opt, with_labels=True,
                    fig_dir="fig"):
        from .utils.plotting import plot_routes as _pr
        return _pr(self, self._env, link_delays, node_delays, opt,
                   fig_dir=fig_dir, with_labels=with_labels)

    def plot_metrics(self, opt, T=None, seed=0, fig_dir="fig"):
        """Run the per-timeslot tracer over the current flows and save the
        reference metrics figure (offloading_v3.py:588-607)."""
        from .sim.timeslot import simulate
        from .utils.plotting import plot_metrics as _pm
        _, _, trace = simulate(self, self._env.jobs, self._env.flows,
                               T=int(T or self.T), seed=seed, trace=True)
        _pm(trace, self, opt, fig_dir=fig_dir)
        return (trace["arrivals"], trace["pkts_in_network"],
                trace["departures"])
