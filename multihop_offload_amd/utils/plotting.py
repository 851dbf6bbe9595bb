"""Network/route visualisation — rebuild of ``util.py:53-98`` (vis_network,
vis_edges) and ``offloading_v3.py:552-586`` (plot_routes)."""

from __future__ import annotations

import os

import numpy as np


def vis_edges(graph, pos, edge_labels, ax=None, font_size=12):
    import networkx as nx
    nx.draw_networkx_edge_labels(graph, pos=pos, edge_labels=edge_labels,
                                 ax=ax, font_size=font_size)


def vis_network(graph, src_nodes, dst_nodes, pos, weights=None, delays=None,
                with_labels=True, ax=None, colors=("g", "r", "b"),
                alpha=1.0):
    """Mobiles as red diamonds, servers as blue squares, edge width by
    weight, node size by delay (reference ``util.py:56-97``)."""
    import networkx as nx
    n = graph.number_of_nodes()
    node_colors = ["y"] * n
    node_sizes = [300.0] * n
    edge_colors = ["k"] * (len(weights) if weights is not None else
                           graph.number_of_edges())
    if weights is not None:
        for i, w in enumerate(weights):
            if w > 0.99:
                edge_colors[i] = colors[0]
    if delays is not None:
        node_sizes = ((np.asarray(delays) / 5.0) ** 2 + 20).tolist()
    for s in src_nodes:
        node_colors[s] = colors[1]
        node_sizes[s] = max(node_sizes[s], 200)
    for d in dst_nodes:
        node_colors[d] = colors[2]
        node_sizes[d] = 200
    nx.draw(graph, node_color=node_colors, node_size=node_sizes,
            with_labels=with_labels, pos=pos, width=weights, ax=ax,
            edge_color=edge_colors, alpha=alpha)


def plot_routes(case, env, link_delays, node_delays, opt, fig_dir="fig",
                with_labels=True):
    """Reference ``AdhocCloud.plot_routes`` (offloading_v3.py:552-586):
    route-weighted edges, delay-sized nodes, saved PNG."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    import networkx as nx

    delay_f = np.nan_to_num(link_delays).sum(axis=1)
    delay_s = np.nan_to_num(node_delays).sum(axis=1) * 100
    g = nx.from_numpy_array(case.adj.astype(float))
    pos = (dict(enumerate(np.asarray(case.pos)))
           if case.pos is not None else nx.spring_layout(g, seed=case.seed))
    mobile_nodes = [f.src for f in env.flows]
    weights = delay_f / 10 + 1
    vis_network(g, mobile_nodes, case.servers, pos, weights, delay_s,
                with_labels)
    os.makedirs(fig_dir, exist_ok=True)
    name = os.path.join(
        fig_dir, "offloading_flow_routes_visual_seed_{}_nodes_{}_{}"
        "_cf{:.1f}_opt{}.png".format(case.seed, case.num_nodes, case.gtype,
                                     case.cf_radius, opt))
    plt.savefig(name, dpi=300, bbox_inches="tight")
    plt.close()
    return name


def plot_metrics(trace, case, opt, fig_dir="fig"):
    """Reference ``AdhocCloud.plot_metrics`` (offloading_v3.py:588-607):
    per-timeslot exogenous arrivals, sink departures, and packets in the
    network, from a ``sim.timeslot.simulate(..., trace=True)`` trace."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    plt.plot(trace["arrivals"])
    plt.plot(trace["departures"])
    plt.plot(trace["pkts_in_network"])
    plt.suptitle("Departures, Arrivals, and Current amount pkts in network")
    plt.xlabel("T")
    plt.ylabel("the number of packages")
    plt.legend(["Exogenous arrivals", "Sink departures", "Pkts in network"],
               loc="upper right")
    os.makedirs(fig_dir, exist_ok=True)
    name = os.path.join(
        fig_dir, "flow_packets_arrivals_per_timeslot_seed_{}_nodes_{}_{}"
        "_cf{:.1f}_opt_{}.png".format(case.seed, case.num_nodes, case.gtype,
                                      case.cf_radius, opt))
    plt.savefig(name, dpi=300)
    plt.close()
    return name
