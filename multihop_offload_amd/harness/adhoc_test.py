"""Test harness — CLI-compatible rebuild of ``src/AdHoc_test.py``.

Fixed file order, methods [baseline, local, GNN]; like the reference, the GNN
branch runs ``forward_backward`` (gradients computed + memorised, never
applied — ``AdHoc_test.py:150-154``), so its runtime includes backward cost.
Writes ``Adhoc_test_data_<set>_load_<s>_T_<T>.csv`` with the reference schema.
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

TEST_COLUMNS = ["filename", "seed", "num_nodes", "m", "num_mobile",
                "num_servers", "num_relays", "num_jobs", "n_instance",
                "Algo", "runtime", "tau", "congest_jobs", "gnn_bl_ratio",
                "gap_2_bl"]


def main(argv=None):
    args = common.build_parser().parse_args(argv)
    rng = np.random.RandomState(args.seed) if args.seed is not None else None

    agent = common.agent_from_args(args, memory_size=1000)
    os.makedirs(args.out, exist_ok=True)
    actor_model = model_dir(args.model_root, args.training_set)
    try:
        agent.load(actor_model)
    except Exception:
        print(f"unable to load {actor_model}")

    names = common.list_cases(args.datapath, args.limit_cases,
                              getattr(args, 'shard', ''))
    output_csv = os.path.join(
        args.out, "Adhoc_test_data_{}_load_{:.2f}_T_{}.csv".format(
            os.path.basename(os.path.normpath(args.datapath)),
            args.arrival_scale, args.T))
    rows = []

    for fid in range(len(names)):
        filepath = os.path.join(args.datapath, names[fid])
        g = common.load_case(filepath, args.T, rng)
        env = AdhocCloudEnv(g)
        runner = (common.EngineRunner(agent, g, seed=args.seed or 0)
                  if args.engine else None)
        t_case = time.time()
        for ni in range(args.instances):
            jobs = JobInstance.sample(g.mobile_nodes, args.arrival_scale,
                                      rng or np.random)
            delay_dict = {}
            for method in ["baseline", "local", "GNN"]:
                env.set_jobs(jobs)
                t0 = time.time()
                delay_emp = (runner.run_method(method, jobs, 0.0)
                             if runner is not None else
                             common.run_method(method, agent, env, 0.0, rng))
                runtime = time.time() - t0
                delay_dict[method] = delay_emp
                rows.append({
                    "filename": names[fid], "seed": g.seed, "n_instance": ni,
                    "num_nodes": g.num_nodes, "m": g.m,
                    "num_servers": len(g.servers),
                    "num_relays": len(g.relays),
                    "num_mobile": (g.num_nodes - len(g.servers)
                                   - len(g.relays)),
                    "num_jobs": jobs.num_jobs, "Algo": method,
                    "runtime": runtime,
                    "tau": np.nanmean(delay_emp),
                    "congest_jobs": int(np.count_nonzero(
                        delay_emp > float(g.T))),
                    "gnn_bl_ratio": np.nanmean(delay_emp
                                               / delay_dict["baseline"]),
                    "gap_2_bl": np.nanmean(delay_emp
                                           - delay_dict["baseline"]),
                })
        print("Runtime {:.3f}s  for network of {} nodes, {} servers, "
              "{} relays".format(time.time() - t_case, g.num_nodes,
                                 len(g.servers), len(g.relays)))
        pd.DataFrame(rows, columns=TEST_COLUMNS).to_csv(output_csv,
                                                        index=False)
    return output_csv


if __name__ == "__main__":
    main()
