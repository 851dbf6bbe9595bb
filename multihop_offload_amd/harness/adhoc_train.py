"""Training harness — CLI-compatible rebuild of ``src/AdHoc_train.py``.

Epoch loop over shuffled .mat cases × ``--instances`` random job instances;
methods [baseline, local, GNN, GNN-test] per instance; replay + ε/explore
decay; per-epoch checkpoint into the ``model_ChebConv_*`` layout; appends
metrics to the reference CSV schema (``AdHoc_train.py:42-47``).
"""

from __future__ import annotations

import os
import time

import numpy as np
import pandas as pd

from ..graphs import JobInstance
from ..env import AdhocCloudEnv
from ..utils.checkpoint import model_dir
from . import common

TRAIN_COLUMNS = ["fid", "filename", "seed", "num_nodes", "m", "num_mobile",
                 "num_servers", "num_relays", "num_jobs", "n_instance",
                 "runtime", "gap_2_bl", "gnn_bl_ratio", "tau",
                 "congest_jobs", "method"]


def main(argv=None):
    args = common.build_parser().parse_args(argv)
    rng = np.random.RandomState(args.seed) if args.seed is not None else None

    agent = common.agent_from_args(args, memory_size=5000)
    os.makedirs(args.out, exist_ok=True)
    actor_model = model_dir(args.model_root, args.training_set)
    try:
        agent.load(actor_model)
    except Exception:
        print(f"unable to load {actor_model}")

    names = common.list_cases(args.datapath, args.limit_cases,
                              getattr(args, 'shard', ''))
    output_csv = os.path.join(
        args.out, "aco_training_data_{}_load_{:.2f}_T_{}.csv".format(
            os.path.basename(os.path.normpath(args.datapath)),
            args.arrival_scale, args.T))
    rows = []
    gidx, losses = 0, []
    explore, explore_decay = 0.1, 0.99
    perm_rng = rng or np.random

    for epoch in range(args.epochs):
        for fid in perm_rng.permutation(len(names)):
            filepath = os.path.join(args.datapath, names[fid])
            g = common.load_case(filepath, args.T, rng)
            env = AdhocCloudEnv(g)
            runner = (common.EngineRunner(agent, g, seed=args.seed or 0)
                      if args.engine else None)
            for ni in range(args.instances):
                jobs = JobInstance.sample(g.mobile_nodes, args.arrival_scale,
                                          rng or np.random)
                delay_dict = {}
                for method in ["baseline", "local", "GNN", "GNN-test"]:
                    env.set_jobs(jobs)
                    t0 = time.time()
                    delay_emp = (runner.run_method(method, jobs, explore)
                                 if runner is not None else
                                 common.run_method(method, agent, env,
                                                   explore, rng))
                    runtime = time.time() - t0
                    delay_dict[method] = delay_emp
                    rows.append({
                        "fid": gidx, "filename": names[fid], "seed": g.seed,
                        "n_instance": ni, "num_nodes": g.num_nodes, "m": g.m,
                        "num_servers": len(g.servers),
                        "num_relays": len(g.relays),
                        "num_mobile": (g.num_nodes - len(g.servers)
                                       - len(g.relays)),
                        "num_jobs": jobs.num_jobs, "method": method,
                        "runtime": runtime,
                        "gap_2_bl": np.nanmean(delay_emp
                                               - delay_dict["baseline"]),
                        "gnn_bl_ratio": np.nanmean(delay_emp
                                                   / delay_dict["baseline"]),
                        "tau": np.nanmean(delay_emp),
                        "congest_jobs": int(np.count_nonzero(
                            delay_emp > float(g.T))),
                    })

            loss = agent.replay(args.batch)
            losses.append(loss)
            print("{} Loss: {:.2f}, explore: {:.4f}".format(
                gidx, np.nanmean(losses), explore))
            if not np.isnan(loss):
                agent.save(os.path.join(actor_model,
                                        f"cp-{epoch:04d}.ckpt"))
                explore = float(np.clip(explore * explore_decay, 0.0, 1.0))
                losses = []
            gidx += 1
            pd.DataFrame(rows, columns=TRAIN_COLUMNS).to_csv(
                output_csv, index=False)
    return output_csv


if __name__ == "__main__":
    main()
