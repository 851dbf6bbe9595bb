"""Dataset generator — rebuild of ``src/data_generation_offloading.py``.

Reproduces the schema and role/bandwidth distributions (the shipped file is
broken as-is: stale ``from offloading import *`` and removed
``nx.from_numpy_matrix`` — see SURVEY.md §3.3):

  * 10 graph sizes 20..110 × ``size`` seeds, BA(m=2) by default
    (er/ws/grp/poisson supported);
  * relays = minimum node cut; Stoer–Wagner min cut splits the rest into a
    server side (smaller partition; Pareto(2)+1 ×100 bandwidths, sorted
    descending) and a mobile side (Pareto(2)+1 ×8);
  * link rates ~ U(30, 70);
  * one documented deviation: leftover server-side nodes get mobile
    Pareto(2)×8 bandwidths instead of the reference's role-0/bandwidth-0
    (see the inline comment in :func:`generate_case`);
  * saved as ``aco_case_seed{S}_m{M}_n{N}_s{num_servers}.mat`` with keys
    {network{num_nodes,seed,m,gtype}, adj (sparse float), link_rate,
    nodes_info (N×2), pos_c (N×2)} (``data_generation_offloading.py:136-144``).
"""

from __future__ import annotations

import argparse
import os

import numpy as np
import scipy.io as sio
import scipy.sparse as sp

from .graphs import build_connectivity

GRAPH_SIZES = [20, 30, 40, 50, 60, 70, 80, 90, 100, 110]


def generate_case(num_nodes: int, seed: int, gtype: str = "ba", m: int = 2,
                  rng=None):
    """Build one case; returns the dict of .mat arrays."""
    import networkx as nx
    rng = rng or np.random

    if gtype == "poisson":
        mm = 3
        while True:
            mm += 1
            adj, pos = build_connectivity(num_nodes, "poisson", seed, mm)
            if nx.is_connected(nx.from_numpy_array(adj)):
                m = mm
                break
    else:
        adj, pos = build_connectivity(num_nodes, gtype, seed, m)
        if pos is None:
            g = nx.from_numpy_array(adj)
            layout = nx.spring_layout(g, seed=seed)
            pos = np.array([layout[i] for i in range(num_nodes)])

    graph = nx.from_numpy_array(adj)
    num_links = graph.number_of_edges()
    jobs_perc = rng.randint(15, 40)
    server_perc = rng.randint(10, 25)
    num_servers = round(server_perc / 100 * num_nodes)
    link_rates = rng.uniform(30, 70, size=(num_links,))

    relay_set = set(nx.minimum_node_cut(graph))
    _, partition = nx.stoer_wagner(graph)
    nodes_info = np.zeros((num_nodes, 2))
    for idx in relay_set:
        nodes_info[idx, 0] = 2
        nodes_info[idx, 1] = 0

    p0 = rng.permutation(list(set(partition[0]) - relay_set)).tolist()
    p1 = rng.permutation(list(set(partition[1]) - relay_set)).tolist()
    partition = (p0, p1)
    server_side = 1 if len(p0) >= len(p1) else 0

    for sidx in range(2):
        part = partition[sidx]
        if sidx == server_side:
            k = min(num_servers, len(part))
            bws = np.flip(np.sort((rng.pareto(2.0, k) + 1) * 100))
            for i in range(k):
                nodes_info[part[i], 0] = 1
                nodes_info[part[i], 1] = bws[i]
            # DELIBERATE DEVIATION from the reference
            # (data_generation_offloading.py:90-133): leftover nodes on the
            # server side (when num_servers < side size) become mobiles
            # with Pareto(2)×8 bandwidths; the reference leaves them as
            # role-0 nodes with bandwidth 0, which makes them job sources
            # that can never compute anything.  Cross-comparisons against
            # reference-GENERATED datasets should account for this shifted
            # role/bandwidth distribution (reference-SHIPPED `.mat` files
            # load unchanged through CaseGraph.from_mat).
            n_rest = len(part) - k
            if n_rest > 0:
                mb = (rng.pareto(2.0, n_rest) + 1) * 8
                for i in range(k, len(part)):
                    nodes_info[part[i], 0] = 0
                    nodes_info[part[i], 1] = mb[i - k]
        else:
            n_near = min(max(0, num_servers - len(partition[server_side])),
                         len(part))
            if n_near > 0:
                bws = (rng.pareto(2.0, n_near) + 1) * 100
                for i in range(n_near):
                    nodes_info[part[i], 0] = 1
                    nodes_info[part[i], 1] = bws[i]
            n_mob = len(part) - n_near
            mb = (rng.pareto(2.0, n_mob) + 1) * 8
            for i in range(n_near, len(part)):
                nodes_info[part[i], 0] = 0
                nodes_info[part[i], 1] = mb[i - n_near]

    # guarantee at least one job source: tiny graphs with a large relay
    # cut can otherwise end up all-server/all-relay
    if not np.any(nodes_info[:, 0] == 0):
        servers = np.nonzero(nodes_info[:, 0] == 1)[0]
        demote = servers[np.argmin(nodes_info[servers, 1])]
        nodes_info[demote, 0] = 0
        nodes_info[demote, 1] = (rng.pareto(2.0) + 1) * 8

    return {
        "network": {"num_nodes": num_nodes, "seed": seed, "m": m,
                    "gtype": gtype},
        "adj": sp.csr_matrix(adj.astype(float)),
        "link_rate": link_rates,
        "nodes_info": nodes_info,
        "pos_c": np.asarray(pos, dtype=float),
    }, num_servers


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--datapath", default="data/aco_data_ba_100", type=str)
    p.add_argument("--gtype", default="ba", type=str)
    p.add_argument("--size", default=100, type=int)
    p.add_argument("--seed", default=500, type=int)
    p.add_argument("--sizes", default=None, type=str,
                   help="comma-separated node counts (default 20..110)")
    args = p.parse_args(argv)
    os.makedirs(args.datapath, exist_ok=True)
    sizes = ([int(s) for s in args.sizes.split(",")] if args.sizes
             else GRAPH_SIZES)
    for i in range(args.size):
        seed = i + args.seed
        for n in sizes:
            case, num_servers = generate_case(n, seed, args.gtype)
            fname = "aco_case_seed{}_m{}_n{}_s{}.mat".format(
                seed, case["network"]["m"], n, num_servers)
            sio.savemat(os.path.join(args.datapath, fname), case)
    print(f"wrote {args.size * len(sizes)} cases to {args.datapath}")


if __name__ == "__main__":
    main()
