"""Working rebuild of the reference demo entry point
(``offloading_v3.main()``, offloading_v3.py:609-686).

The reference's demo is stale and crashes as shipped (calls ``offloading``
without ``hpmtx``, unpacks 2 of 3 ``run()`` returns, and ``plot_metrics``
reads per-timeslot arrays nothing assigns — SURVEY.md §3.4).  This is the
same scenario made functional: a 15-node BA network with 5 servers and
2 relays, 5 Poisson job flows, greedy-baseline offloading, the analytic
evaluation, the route figure, and (``--trace``) the per-timeslot packet
simulation behind the metrics figure.
"""

from __future__ import annotations

import argparse
import os
import time

import numpy as np

from ..env import AdhocCloud, apsp


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("opt", nargs="?", type=int, default=0)
    p.add_argument("--fig_dir", type=str, default="fig")
    p.add_argument("--out", type=str, default="out")
    p.add_argument("--trace", action="store_true",
                   help="also run the per-timeslot simulator + metrics plot")
    p.add_argument("--seed", type=int, default=3)
    args = p.parse_args(argv)

    NUM_NODES, T, link_rate, cf_radius = 15, 500, 50, 0.0
    arrival_const = 2.0

    os.makedirs(args.fig_dir, exist_ok=True)
    os.makedirs(args.out, exist_ok=True)

    t0 = time.time()
    net = AdhocCloud(NUM_NODES, T, args.seed, cf_radius=cf_radius, trace=True)
    for s, bw in [(4, 300), (14, 300), (13, 300), (12, 300), (1, 200)]:
        net.add_server(s, proc_bw=bw)
    net.add_relay(3)
    net.add_relay(0)
    for src, rate in [(10, 0.05), (11, 0.03), (7, 0.02), (8, 0.02), (6, 0.05)]:
        net.add_job(src, rate=arrival_const * rate)
    net.links_init(link_rate, rng=np.random.RandomState(args.seed))

    _, dlist_bl, dproc_bl = net.dmtx_baseline()
    sp_baseline = apsp(net, dlist_bl)
    np.fill_diagonal(sp_baseline, np.where(dproc_bl > 0, dproc_bl, T))

    decisions, delay_est = net.offloading(sp_baseline)

    logfile = os.path.join(
        args.out, "Output_seed_{}_nodes_{}_opt_{}.txt".format(
            args.seed, NUM_NODES, args.opt))
    with open(logfile, "a") as f:
        print("Edges:", file=f)
        print(net.edges.tolist(), file=f)
        print("Link Rates:", file=f)
        print(net.link_rates, file=f)

    print("Init graph in {:.3f} seconds: conflict radius {}, degree {:.2f}"
          .format(time.time() - t0, net.cf_radius, net.mean_conflict_degree))

    t0 = time.time()
    delay_links, delay_nodes, _ = net.run()
    print("Evaluation of {} time slots in {:.3f} seconds"
          .format(T, time.time() - t0))
    net.plot_routes(delay_links, delay_nodes, args.opt, fig_dir=args.fig_dir)

    delay_emp = (np.nansum(delay_links, axis=0)
                 + np.nansum(delay_nodes, axis=0))
    print("Delay estimation: {}\nDelay empirical: {}".format(
        delay_est, delay_emp))
    if args.trace:
        net.plot_metrics(args.opt, fig_dir=args.fig_dir)
    print("Done")
    return decisions, delay_est, delay_emp


if __name__ == "__main__":
    main()
