"""Per-timeslot queueing simulator — validation oracle for the analytic
evaluator.

The reference's product path is the analytic M/M/1 + contention fixed-point
evaluator (``offloading_v3.py:455-550``); its repo carries vestiges of an
older per-timeslot packet simulator.  This module is a
compact discrete-time simulator used ONLY as a statistical cross-check of
the analytic evaluator in tests:

  * tasks arrive per job as Poisson(rate) per slot;
  * each task carries ``ul`` units to its destination hop-by-hop, is
    processed at the server (``ul`` units at the node bandwidth), and
    returns ``dl`` units hop-by-hop;
  * each link serves its FIFO with capacity ``rate_l / (1 + #busy
    conflicting links)`` units per slot (the same contention-sharing model
    the fixed point solves in steady state); servers serve at ``proc_bw``.

Under stable load the simulated mean task sojourn should track the analytic
per-job delays within a small factor (M/D/1-vs-M/M/1-style gap).
"""

from __future__ import annotations

from typing import List

import numpy as np

from ..env import AdhocCloudEnv, Flow
from ..graphs import CaseGraph, JobInstance


class _Task:
    __slots__ = ("job", "stage", "remaining", "t0", "done_at")

    def __init__(self, job: int, t0: int):
        self.job = job
        self.stage = 0           # index into the task's leg sequence
        self.remaining = 0.0
        self.t0 = t0
        self.done_at = -1


def simulate(g: CaseGraph, jobs: JobInstance, flows: List[Flow],
             T: int = 2000, seed: int = 0, warmup: int = 0,
             trace: bool = False):
    """Returns (mean_delay_per_job (J,), completed_counts (J,)); with
    ``trace=True`` also a dict of per-slot totals (arrivals, sink
    departures, packets in network — the reference ``plot_metrics``
    series, offloading_v3.py:588-607)."""
    rng = np.random.RandomState(seed)
    E, J = g.num_links, jobs.num_jobs

    # per-job leg sequence: [("link", l, ul)...] , ("node", dst, ul),
    # [("link", l, dl)... reversed]
    legs = []
    env = AdhocCloudEnv(g)
    for j, f in enumerate(flows):
        links = env.route_links(f) if f.src != f.dst else np.empty(0, int)
        seq = [("link", int(l), float(jobs.ul[j])) for l in links]
        seq.append(("node", int(f.dst), float(jobs.ul[j])))
        seq += [("link", int(l), float(jobs.dl[j])) for l in links[::-1]]
        legs.append(seq)

    link_q: List[List[_Task]] = [[] for _ in range(E)]
    node_q: List[List[_Task]] = [[] for _ in range(g.num_nodes)]
    delays = [[] for _ in range(J)]
    tr_arr = np.zeros(T, dtype=np.int64)
    tr_dep = np.zeros(T, dtype=np.int64)
    tr_net = np.zeros(T, dtype=np.int64)
    in_network = 0

    def enqueue(task: _Task, t: int):
        while task.stage < len(legs[task.job]):
            kind, idx, units = legs[task.job][task.stage]
            task.remaining = units
            (link_q[idx] if kind == "link" else node_q[idx]).append(task)
            return
        task.done_at = t
        if task.t0 >= warmup:
            delays[task.job].append(t - task.t0)

    for t in range(T):
        # arrivals
        for j in range(J):
            for _ in range(rng.poisson(jobs.rates[j])):
                task = _Task(j, t)
                tr_arr[t] += 1
                in_network += 1
                enqueue(task, t)
                if task.done_at >= 0:           # zero-length leg sequence
                    in_network -= 1
                    tr_dep[t] += 1
        # link service with contention sharing
        busy = np.array([1 if q else 0 for q in link_q])
        nb_busy = np.zeros(E)
        rows = np.repeat(np.arange(E), np.diff(g.conf_indptr))
        np.add.at(nb_busy, rows, busy[g.conf_indices])
        finished = []
        for l in range(E):
            if not link_q[l]:
                continue
            cap = g.link_rates[l] / (1.0 + nb_busy[l])
            while cap > 0 and link_q[l]:
                head = link_q[l][0]
                served = min(cap, head.remaining)
                head.remaining -= served
                cap -= served
                if head.remaining <= 1e-12:
                    link_q[l].pop(0)
                    head.stage += 1
                    finished.append(head)
                else:
                    break
        for n in range(g.num_nodes):
            if not node_q[n]:
                continue
            cap = g.proc_bws[n]
            while cap > 0 and node_q[n]:
                head = node_q[n][0]
                served = min(cap, head.remaining)
                head.remaining -= served
                cap -= served
                if head.remaining <= 1e-12:
                    node_q[n].pop(0)
                    head.stage += 1
                    finished.append(head)
                else:
                    break
        for task in finished:
            enqueue(task, t + 1)
            if task.done_at >= 0:
                in_network -= 1
                tr_dep[min(t + 1, T - 1)] += 1
        tr_net[t] = in_network

    mean_delay = np.array([np.mean(d) if d else np.nan for d in delays])
    counts = np.array([len(d) for d in delays])
    if trace:
        return mean_delay, counts, {"arrivals": tr_arr, "departures": tr_dep,
                                    "pkts_in_network": tr_net}
    return mean_delay, counts
