"""Shared harness plumbing: flags, case loading, the per-method episode.

Flag names/defaults mirror ``gnn_offloading_agent.py:42-60``; method branches
mirror ``AdHoc_train.py:124-157`` / ``AdHoc_test.py:115-156``.
"""

from __future__ import annotations

import argparse
import os
from typing import Optional

import numpy as np

from ..agent import ACOAgent, AgentConfig
from ..env import AdhocCloudEnv, apsp, delay_empirical
from ..graphs import CaseGraph


def build_parser(default_datapath="../data_100") -> argparse.ArgumentParser:
    p = argparse.ArgumentParser()
    add = p.add_argument
    add("--datapath", type=str, default=default_datapath, help="input data path")
    add("--out", type=str, default="out", help="output data path")
    add("--T", type=int, default=1000)
    add("--prob", action="store_true", help="probabilistic decision")
    add("--engine", action="store_true",
        help="run all method branches through the batched device engine "
             "(B=1 per case) instead of the numpy oracle — the fused HIP "
             "path on GPU; identical numbers at explore=0")
    add("--training_set", type=str, default="BAm2")
    add("--learning_rate", type=float, default=0.0001)
    add("--learning_decay", type=float, default=1.0)
    add("--arrival_scale", type=float, default=0.1)
    add("--epochs", type=int, default=201)
    add("--num_layer", type=int, default=5)
    add("--dropout", type=float, default=0.0)
    add("--weight_decay", type=float, default=5e-4)
    add("--epsilon", type=float, default=1.0)
    add("--epsilon_min", type=float, default=0.001)
    add("--epsilon_decay", type=float, default=0.985)
    add("--gamma", type=float, default=1.0)
    add("--batch", type=int, default=100)
    # MI355X-native knobs
    add("--K", type=int, default=2, help="Chebyshev order")
    add("--device", type=str, default="cpu")
    add("--dtype", type=str, default=None,
        help="float64 (CPU default) or float32 (GPU default)")
    add("--model_root", type=str, default="model")
    add("--instances", type=int, default=10)
    add("--seed", type=int, default=None, help="RNG seed (unseeded like the "
        "reference when omitted)")
    add("--limit_cases", type=int, default=0,
        help="process at most this many .mat cases (0 = all)")
    add("--shard", type=str, default="",
        help="'I:K' — process cases I::K only (parallel sharding of a "
             "large run; merge the per-shard CSVs afterwards)")
    return p


def agent_from_args(args, memory_size: int) -> ACOAgent:
    dtype = args.dtype or ("float32" if args.device.startswith("cuda")
                           else "float64")
    cfg = AgentConfig(
        T=args.T, prob=args.prob, learning_rate=args.learning_rate,
        learning_decay=args.learning_decay, arrival_scale=args.arrival_scale,
        epochs=args.epochs, num_layer=args.num_layer, dropout=args.dropout,
        weight_decay=args.weight_decay, epsilon=args.epsilon,
        epsilon_min=args.epsilon_min, epsilon_decay=args.epsilon_decay,
        gamma=args.gamma, batch=args.batch, K=args.K, device=args.device,
        dtype=dtype, seed=args.seed)
    return ACOAgent(cfg, memory_size)


def load_case(filepath: str, T: int,
              rng: Optional[np.random.RandomState] = None) -> CaseGraph:
    """Load a .mat case and initialise link rates (``AdHoc_train.py:85-110``)."""
    g = CaseGraph.from_mat(filepath, t_max=T)
    g.links_init(g.mat_link_rate, rng=rng or np.random)
    return g


def run_method(method: str, agent: ACOAgent, env: AdhocCloudEnv,
               explore: float = 0.0,
               rng: Optional[np.random.RandomState] = None):
    """One method branch; returns per-job empirical delays.
    Methods: baseline / local / GNN (with backward) / GNN-test (fwd only)."""
    g = env.g
    if method == "baseline":
        dmtx_bl, dlist_bl, dproc_bl = env.dmtx_baseline()
        dproc_bl = dproc_bl.copy()
        dproc_bl[dproc_bl <= 0] = float(g.T)
        dlist_bl = np.where(dlist_bl > 0, dlist_bl, float(g.T))
        sp_bl = apsp(g, dlist_bl)
        np.fill_diagonal(sp_bl, dproc_bl)
        env.offloading(sp_bl, g.sp_hop, rng=rng)
        dl_links, dl_nodes, _ = env.run()
    elif method == "local":
        _, _, dproc_bl = env.dmtx_baseline()
        env.local_compute(dproc_bl)
        dl_links, dl_nodes, _ = env.run()
    elif method == "GNN":
        out = agent.forward_backward(env, explore, rng)
        dl_links, dl_nodes = out[1], out[2]
    elif method == "GNN-test":
        dl_links, dl_nodes, _ = agent.forward_env(env, rng)
    else:
        raise ValueError(method)
    return delay_empirical(dl_links, dl_nodes)


def list_cases(datapath: str, limit: int = 0, shard: str = ""):
    names = sorted(os.listdir(datapath))
    names = [n for n in names if n.endswith(".mat")]
    if limit:
        names = names[:limit]
    if shard:
        i, k = (int(x) for x in shard.split(":"))
        names = names[i::k]
    return names


class EngineRunner:
    """Engine-backed method execution for the compat harnesses
    (``--engine``): one B=1 :class:`EpisodeEngine` per case visit, so the
    reference workflow runs on the fused HIP path on GPU instead of the
    numpy oracle.  Numerically identical to ``run_method`` at explore=0
    (test_engine's B=1 oracle equivalences); with explore>0 the decision
    RNG is the device generator rather than numpy, so individual draws
    differ while the statistics match.  The GNN branch memorises the
    per-instance gradient set exactly like ``ACOAgent.forward_backward``
    (B=1: the summed engine gradient IS the instance gradient)."""

    def __init__(self, agent: ACOAgent, g, seed: int = 0):
        import torch
        from ..engine import EpisodeEngine
        self.agent = agent
        self.engine = EpisodeEngine([g], agent.model,
                                    device=str(agent.device),
                                    dtype=agent.dtype)
        self.gen = torch.Generator(device=str(agent.device))
        self.gen.manual_seed(seed * 9973 + g.seed)
        # per-case kernel-RNG seed: without this every case engine would
        # replay the same in-kernel ε-greedy stream (correlated explore)
        self.engine.set_rng_seed(seed * 9973 + g.seed)

    def run_method(self, method: str, jobs, explore: float = 0.0):
        import torch
        eng = self.engine
        jb = eng.pack_jobs([jobs])
        if method == "baseline":
            res = eng.baseline_episode(jb)
        elif method == "local":
            res = eng.local_episode(jb)
        elif method == "GNN":
            for p in eng.model.parameters():
                p.grad = None
            res = eng.gnn_episode(jb, explore=explore, gen=self.gen,
                                  train=True, prob=self.agent.cfg.prob)
            grads = [p.grad.detach().clone() if p.grad is not None
                     else torch.zeros_like(p)
                     for p in eng.model.parameters()]
            self.agent.memorize(grads, float(res.loss_fn),
                                float(res.loss_mse))
        elif method == "GNN-test":
            res = eng.gnn_episode(jb, train=False)
        else:
            raise ValueError(method)
        k = jobs.num_jobs
        return res.delay_emp[0, :k].detach().cpu().numpy()
