"""Graph data layer: canonical tensor representation of one network case.

Reference analog: the graph-construction half of ``AdhocCloud``
(``/root/reference/src/offloading_v3.py:30-78,193-260``) and
``graph_expand`` (``offloading_v3.py:262-339``).

Design differences (MI355X-first, not a port):

* **Canonical link order** — links are the lexicographically sorted edges
  ``(u, v), u < v`` of the connectivity graph.  The extended ("self-loop
  expanded") edge set is ``[real links 0..E-1] ++ [virtual self-loop edges
  for non-relay nodes in ascending node order]``, so the reference's
  ``maps_ol_el`` is ``arange(E)`` by construction and every ``list.index``
  scan (``offloading_v3.py:299-331``) disappears.
* Everything is a flat numpy/torch tensor (CSR neighbor lists, CSR conflict
  lists, index maps) built once per case on the host; the per-step compute
  consumes only tensors on device.
"""

from __future__ import annotations

import dataclasses
from typing import Optional

import numpy as np


# ---------------------------------------------------------------------------
# construction helpers
# ---------------------------------------------------------------------------

def _edges_from_adj(adj: np.ndarray) -> np.ndarray:
    """Upper-triangle edge list (lexicographic) from a dense 0/1 adjacency."""
    iu, ju = np.nonzero(np.triu(adj, k=1))
    return np.stack([iu, ju], axis=1).astype(np.int64)


def build_connectivity(num_nodes: int, gtype: str, seed: int, m: int = 2,
                       pos: Optional[np.ndarray] = None):
    """Generate the connectivity graph. Same graph families and parameters as
    the reference (``offloading_v3.py:39-59``): ba / grp / ws / er, plus
    'poisson' (``data_generation_offloading.py:34-50``).

    Returns (adj dense int8, pos or None).
    """
    import networkx as nx

    gtype = gtype.lower()
    if gtype == "ba":
        g = nx.barabasi_albert_graph(num_nodes, m, seed=seed)
    elif gtype == "grp":
        g = nx.gaussian_random_partition_graph(num_nodes, 15, 3, 0.4, 0.2, seed=seed)
    elif gtype == "ws":
        g = nx.connected_watts_strogatz_graph(num_nodes, k=6, p=0.2, seed=seed)
    elif gtype == "er":
        g = nx.fast_gnp_random_graph(num_nodes, 15.0 / float(num_nodes), seed=seed)
    elif gtype == "poisson":
        rng = np.random.RandomState(seed)
        density = float(m) / np.pi
        side = np.sqrt(float(num_nodes) / density)
        xys = rng.uniform(0, side, (num_nodes, 2))
        d = np.sqrt(((xys[:, None, :] - xys[None, :, :]) ** 2).sum(-1))
        adj = (d <= 1.0).astype(np.int8)
        np.fill_diagonal(adj, 0)
        return adj, xys
    else:
        raise ValueError(f"unsupported graph model: {gtype}")
    adj = nx.to_numpy_array(g, nodelist=range(num_nodes), dtype=np.int8)
    return adj, pos


def _csr_from_lists(lists, total):
    indptr = np.zeros(len(lists) + 1, dtype=np.int64)
    for i, l in enumerate(lists):
        indptr[i + 1] = indptr[i] + len(l)
    indices = np.empty(indptr[-1], dtype=np.int64)
    for i, l in enumerate(lists):
        indices[indptr[i]:indptr[i + 1]] = l
    return indptr, indices


@dataclasses.dataclass
class JobInstance:
    """One set of jobs on a case (reference ``Job``, ``offloading_v3.py:131-138``)."""
    sources: np.ndarray        # (J,) int64 source (mobile) node per job
    rates: np.ndarray          # (J,) float arrival rate
    ul: np.ndarray             # (J,) float uplink data size (default 100)
    dl: np.ndarray             # (J,) float downlink data size (default 1)

    @staticmethod
    def sample(mobile_nodes: np.ndarray, arrival_scale: float,
               rng: np.random.RandomState) -> "JobInstance":
        """Job sampling exactly as the harness does it
        (``AdHoc_train.py:112-121``): shuffle mobiles, J ~ U{0.3·M, M-1},
        rates ~ U(0.1, 0.5)·arrival_scale, ul=100, dl=1."""
        mobile = rng.permutation(mobile_nodes)
        num_mobile = mobile.size
        num_jobs = rng.randint(int(0.3 * num_mobile), num_mobile)
        rates = rng.uniform(0.1, 0.5, (num_jobs,)) * arrival_scale
        return JobInstance(
            sources=mobile[:num_jobs].astype(np.int64),
            rates=rates,
            ul=np.full(num_jobs, 100.0),
            dl=np.full(num_jobs, 1.0),
        )

    @property
    def num_jobs(self):
        return len(self.sources)


class CaseGraph:
    """One network case: connectivity + conflict (line) graph + extended graph.

    All index maps the per-step compute needs are precomputed tensors.
    """

    def __init__(self, num_nodes: int, t_max: int = 1000, seed: int = 3, m: int = 2,
                 gtype: str = "ba", adj: Optional[np.ndarray] = None,
                 pos: Optional[np.ndarray] = None, cf_radius: float = 0.0):
        self.num_nodes = int(num_nodes)
        self.T = int(t_max)
        self.seed = int(seed)
        self.m = int(m)
        self.gtype = gtype
        self.cf_radius = float(cf_radius)

        if adj is None:
            adj, pos = build_connectivity(self.num_nodes, gtype, self.seed, self.m, pos)
        adj = np.asarray(adj)
        if adj.shape != (self.num_nodes, self.num_nodes):
            raise ValueError("adjacency shape mismatch")
        self.adj = (adj != 0).astype(np.int8)
        self.pos = pos

        # --- links (canonical lexicographic order) ---
        self.edges = _edges_from_adj(self.adj)          # (E,2) u<v
        self.num_links = len(self.edges)
        E, N = self.num_links, self.num_nodes

        # link id lookup matrix (reference ``link_matrix``, offloading_v3.py:226-241)
        self.link_matrix = -np.ones((N, N), dtype=np.int64)
        self.link_matrix[self.edges[:, 0], self.edges[:, 1]] = np.arange(E)
        self.link_matrix[self.edges[:, 1], self.edges[:, 0]] = np.arange(E)

        # adjacency CSR with per-neighbor link ids (for greedy routing)
        nbr_lists, nbr_link_lists = [], []
        for v in range(N):
            nbs = np.nonzero(self.adj[v])[0]
            nbr_lists.append(nbs)
            nbr_link_lists.append(self.link_matrix[v, nbs])
        self.adj_indptr, self.adj_indices = _csr_from_lists(nbr_lists, None)
        _, self.adj_link_ids = _csr_from_lists(nbr_link_lists, None)

        # --- conflict (line) graph over real links ---
        # two links conflict iff they share an endpoint (offloading_v3.py:65).
        inc = [[] for _ in range(N)]
        for l, (u, v) in enumerate(self.edges):
            inc[u].append(l)
            inc[v].append(l)
        conf_sets = [set() for _ in range(E)]
        for node_links in inc:
            for l in node_links:
                conf_sets[l].update(node_links)
        if self.cf_radius > 0.5:
            self._add_conflict_relations(conf_sets)
        conf_lists = [sorted(s - {l}) for l, s in enumerate(conf_sets)]
        self.conf_indptr, self.conf_indices = _csr_from_lists(conf_lists, None)
        self.cf_degs = np.diff(self.conf_indptr).astype(np.float64)
        self.mean_conflict_degree = float(self.cf_degs.mean()) if E else 0.0

        # --- roles / processing bandwidths (offloading_v3.py:170-184) ---
        self.roles = np.zeros(N, dtype=np.float64)      # 0 mobile, 1 server, 2 relay
        self.proc_bws = 2.0 * np.ones(N, dtype=np.float64)
        self.servers: list = []
        self.relays: list = []

        self.link_rates = np.zeros(E, dtype=np.float64)
        self._ext = None      # lazily built extended-graph bundle
        self._sp_hop = None   # lazily computed static hop-count APSP

    # -- reference-API mutators ------------------------------------------------
    def add_server(self, node: int, proc_bw: float):
        self.roles[node] = 1
        self.proc_bws[node] = proc_bw
        self.servers.append(int(node))
        self._ext = None

    def add_relay(self, node: int):
        self.roles[node] = 2
        self.proc_bws[node] = 0.0
        self.relays.append(int(node))
        self._ext = None

    def set_mobile_bw(self, node: int, proc_bw: float):
        self.roles[node] = 0
        self.proc_bws[node] = proc_bw
        self._ext = None

    def links_init(self, rates, std: float = 2.0, rng: Optional[np.random.RandomState] = None):
        """Gaussian-perturbed, clipped, rounded link rates
        (``offloading_v3.py:252-260``)."""
        rng = rng or np.random
        rates = np.asarray(rates, dtype=np.float64)
        if rates.ndim > 0 and rates.size > 1:
            assert rates.size == self.num_links
        hi = rates + 3 * std
        self.link_rates = np.round(np.clip(
            rng.normal(rates, std, size=(self.num_links,)), 0, hi))

    # -- mobility (offloading_v3.py:80-129) -----------------------------------
    def random_walk(self, ss: float = 0.1, n: int = 10,
                    rng: Optional[np.random.RandomState] = None):
        """Perturb n random node positions by N(0, ss) within the bounding
        box and rebuild the unit-disk connectivity; retries until connected
        (reference ``random_walk``).  Returns (adj, new_pos)."""
        if self.pos is None:
            raise ValueError("random_walk needs node positions")
        rng = rng or np.random
        from scipy.sparse import csr_matrix
        from scipy.sparse.csgraph import connected_components
        pos = np.asarray(self.pos, dtype=np.float64)
        b_min, b_max = pos.min() - 0.05, pos.max() + 0.05
        for _ in range(1000):
            mask = rng.choice(self.num_nodes, size=n, replace=False)
            new_pos = pos.copy()
            new_pos[mask] += rng.normal(0, ss, size=(n, 2))
            new_pos = new_pos.clip(b_min, b_max)
            d = np.sqrt(((new_pos[:, None] - new_pos[None, :]) ** 2).sum(-1))
            adj = (d <= 1.0).astype(np.int8)
            np.fill_diagonal(adj, 0)
            ncomp, _ = connected_components(csr_matrix(adj), directed=False)
            if ncomp == 1:
                return adj, new_pos
        raise RuntimeError("random_walk could not find a connected topology")

    def topology_update(self, adj, pos):
        """Rebuild the topology in place from a new adjacency (reference
        ``topology_update``).  Returns ``new_links_map``: for each new link,
        the old link id carrying the same endpoints, or -1."""
        old_lm = self.link_matrix
        new = CaseGraph(self.num_nodes, t_max=self.T, seed=self.seed,
                        m=self.m, gtype=self.gtype, adj=adj, pos=pos,
                        cf_radius=self.cf_radius)
        new_links_map = old_lm[new.edges[:, 0], new.edges[:, 1]]
        # carry roles/bandwidths over; reset per-link state
        new.roles = self.roles
        new.proc_bws = self.proc_bws
        new.servers = self.servers
        new.relays = self.relays
        self.__dict__.update(new.__dict__)
        return new_links_map

    def clone_with_rates(self, base_rates,
                         rng: Optional[np.random.RandomState] = None,
                         std: float = 2.0) -> "CaseGraph":
        """Cheap replica: shares all topology structure (CSR, maps, hop
        APSP), draws fresh link rates.  Used to build large batches from a
        few distinct topologies without re-running graph construction."""
        import copy as _copy
        g = _copy.copy(self)
        rng = rng or np.random
        base = np.asarray(base_rates, dtype=np.float64)
        g.link_rates = np.round(np.clip(
            rng.normal(base, std, size=(self.num_links,)), 0,
            base + 3 * std))
        if self._ext is not None:
            e = _copy.copy(self._ext)
            e._g = g
            e.edge_rate_ext = e.edge_rate_ext.copy()
            g._ext = e
        return g

    def pad_to(self, n_target: int) -> "CaseGraph":
        """Inert node padding for mixed-size batching in ONE engine: a new
        case with ``n_target - num_nodes`` additional isolated relay nodes
        (no links, no virtual edges, bw 0).  Behaviour-invariant end to
        end: the link set, canonical ordering, conflict CSR and extended
        graph are identical to the original's; padded nodes are
        unreachable (inf APSP rows), never sampled as job sources (not
        mobile), never chosen as servers, and contribute no features.
        This lets batches mix graph sizes without per-size engine buckets
        and without any kernel change."""
        assert n_target >= self.num_nodes
        if n_target == self.num_nodes:
            return self
        N0 = self.num_nodes
        adj = np.zeros((n_target, n_target), dtype=np.int8)
        adj[:N0, :N0] = self.adj
        pos = None
        if self.pos is not None:
            pos = np.vstack([np.asarray(self.pos, dtype=np.float64),
                             np.zeros((n_target - N0, 2))])
        g = CaseGraph(n_target, t_max=self.T, seed=self.seed, m=self.m,
                      gtype=self.gtype, adj=adj, pos=pos,
                      cf_radius=self.cf_radius)
        g.real_n = N0          # kernels bound their loops at the real size
        for v in range(N0):
            if self.roles[v] == 2:
                g.add_relay(v)
            elif self.roles[v] == 1:
                g.add_server(v, self.proc_bws[v])
            else:
                g.set_mobile_bw(v, self.proc_bws[v])
        for v in range(N0, n_target):
            g.add_relay(v)
        g.link_rates = self.link_rates.copy()
        return g

    # -- conflict radius augmentation (offloading_v3.py:193-224) --------------
    def _add_conflict_relations(self, conf_sets):
        if self.pos is None:
            raise ValueError("cf_radius conflicts need node positions")
        pos = np.asarray(self.pos, dtype=np.float64)
        d = np.sqrt(((pos[:, None, :] - pos[None, :, :]) ** 2).sum(-1))
        link_dist = d[self.edges[:, 0], self.edges[:, 1]]
        intf = self.cf_radius * np.nanmedian(link_dist)
        for l, (u, v) in enumerate(self.edges):
            near = np.union1d(np.nonzero(d[u] < intf)[0], np.nonzero(d[v] < intf)[0])
            for w in near:
                for x in np.nonzero(self.adj[w])[0]:
                    ll = self.link_matrix[w, x]
                    if ll >= 0 and ll != l:
                        conf_sets[l].add(int(ll))
                        conf_sets[ll].add(int(l))

    # -- extended graph (reference graph_expand, offloading_v3.py:262-339) ----
    @property
    def ext(self):
        if self._ext is None:
            self._ext = _ExtendedGraph(self)
        return self._ext

    # -- static hop-count APSP -------------------------------------------------
    @property
    def sp_hop(self) -> np.ndarray:
        """Unweighted all-pairs shortest hop counts (static per case).
        Reference recomputes this per step (``util.py:101-110``); it only
        depends on topology, so we compute it once."""
        if self._sp_hop is None:
            from scipy.sparse import csr_matrix
            from scipy.sparse.csgraph import shortest_path
            a = csr_matrix(self.adj.astype(np.float64))
            self._sp_hop = shortest_path(a, method="D", unweighted=True)
        return self._sp_hop

    @property
    def mobile_nodes(self) -> np.ndarray:
        return np.nonzero(self.roles == 0)[0]

    # -- (de)serialisation to the reference .mat schema ------------------------
    @staticmethod
    def from_mat(path: str, t_max: int = 1000, cf_radius: float = 0.0) -> "CaseGraph":
        """Load a case from the reference's .mat schema
        (``data_generation_offloading.py:138-144``) including roles/bws."""
        import scipy.io as sio
        mat = sio.loadmat(path)
        net = mat["network"][0, 0]
        num_nodes = int(np.asarray(net["num_nodes"]).flatten()[0])
        seed = int(np.asarray(net["seed"]).flatten()[0])
        m = int(np.asarray(net["m"]).flatten()[0])
        gtype = str(np.asarray(net["gtype"]).flatten()[0])
        adj = np.asarray(mat["adj"].todense() if hasattr(mat["adj"], "todense")
                         else mat["adj"])
        pos = np.asarray(mat["pos_c"], dtype=np.float64)
        g = CaseGraph(num_nodes, t_max=t_max, seed=seed, m=m, gtype=gtype,
                      adj=adj, pos=pos, cf_radius=cf_radius)
        g.mat_link_rate = np.asarray(mat["link_rate"], dtype=np.float64).flatten()
        nodes_info = np.asarray(mat["nodes_info"], dtype=np.float64)
        for nidx in range(num_nodes):
            role, bw = nodes_info[nidx, 0], float(nodes_info[nidx, 1])
            if role == 2:
                g.add_relay(nidx)
            elif role == 1:
                g.add_server(nidx, bw)
            else:
                g.set_mobile_bw(nidx, bw)
        g.nodes_info = nodes_info
        return g


class _ExtendedGraph:
    """Self-loop-expanded graph: one virtual node per non-relay node, one
    virtual edge (idx, N+idx).  Extended edge ordering: real links first (same
    ids as CaseGraph), then virtual edges for non-relay nodes ascending.

    Fields mirror the reference's ``graph_expand`` output
    (``offloading_v3.py:262-339``) under the canonical ordering:
      * ``maps_ol_el`` == arange(E) (implicit; not materialised)
      * ``maps_on_el[k]`` = extended-edge id of the k-th computing node's
        virtual edge, computing nodes = non-relay nodes ascending
    """

    def __init__(self, g: CaseGraph):
        N, E = g.num_nodes, g.num_links
        comp_nodes = np.nonzero(g.roles < 2)[0]         # non-relay, ascending
        self.comp_nodes = comp_nodes.astype(np.int64)
        self.num_edges_ext = E + len(comp_nodes)
        Ee = self.num_edges_ext

        # virtual-edge id per node (-1 for relays)
        self.node_vedge = -np.ones(N, dtype=np.int64)
        self.node_vedge[comp_nodes] = E + np.arange(len(comp_nodes))
        self.maps_on_el = self.node_vedge[comp_nodes]    # == E + arange

        # per-extended-edge static attributes
        self.edge_self_loop = np.zeros(Ee, dtype=np.float64)
        self.edge_self_loop[E:] = 1.0
        self.edge_as_server = np.zeros(Ee, dtype=np.float64)
        self.edge_as_server[self.node_vedge[np.nonzero(g.roles == 1)[0]]] = 1.0
        self.edge_rate_ext = np.zeros(Ee, dtype=np.float64)
        self.edge_rate_ext[:E] = g.link_rates            # NOTE: rebuilt on demand
        self.edge_rate_ext[E:] = g.proc_bws[comp_nodes]

        # extended line-graph CSR: extended edges conflict iff they share a
        # real node.  Virtual edge (u, N+u) is incident only to real node u.
        inc = [[] for _ in range(N)]
        for l, (u, v) in enumerate(g.edges):
            inc[u].append(l)
            inc[v].append(l)
        for k, u in enumerate(comp_nodes):
            inc[u].append(E + k)
        conf_sets = [set() for _ in range(Ee)]
        for node_links in inc:
            for l in node_links:
                conf_sets[l].update(node_links)
        conf_lists = [sorted(s - {l}) for l, s in enumerate(conf_sets)]
        self.ext_indptr, self.ext_indices = _csr_from_lists(conf_lists, None)
        self._g = g

    def refresh_rates(self):
        """Re-sync edge_rate_ext after links_init / role changes."""
        g = self._g
        self.edge_rate_ext[:g.num_links] = g.link_rates
        self.edge_rate_ext[g.num_links:] = g.proc_bws[self.comp_nodes]

    def jobs_arrivals(self, jobs: JobInstance) -> np.ndarray:
        """Per-extended-edge exogenous arrival feature: sum of rate*ul of the
        jobs sourced at the owning node, placed on that node's virtual edge
        (``offloading_v3.py:277-282,320,328``)."""
        g = self._g
        per_node = np.zeros(g.num_nodes)
        np.add.at(per_node, jobs.sources, jobs.rates * jobs.ul)
        out = np.zeros(self.num_edges_ext)
        mask = self.node_vedge >= 0
        out[self.node_vedge[mask]] = per_node[mask]
        return out

    def features(self, jobs: JobInstance) -> np.ndarray:
        """Ē×4 node features of the extended line graph: [self_loop, rate,
        job_arrival, is_server] (``gnn_offloading_agent.py:218-224``)."""
        self.refresh_rates()
        x = np.zeros((self.num_edges_ext, 4))
        x[:, 0] = self.edge_self_loop
        x[:, 1] = self.edge_rate_ext
        x[:, 2] = self.jobs_arrivals(jobs)
        x[:, 3] = self.edge_as_server
        return x
