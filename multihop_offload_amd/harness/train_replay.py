"""Reference-faithful replay training at engine speed.

``train_batched`` applies the batch-mean gradient per step (the DP-friendly
mode).  This trainer reproduces the REFERENCE's training dynamics exactly
(AdHoc_train.py:112-209 + gnn_offloading_agent.py:141-169) while using the
batched engine for the heavy lifting:

  * every step, a batch of episodes yields ONE gradient set PER INSTANCE
    (``gnn_episode(per_sample=True)``) which is pushed into a replay deque
    (maxlen 5000, like the reference's memory);
  * every ``replay_every`` steps, ``batch`` gradient sets are sampled and
    applied SEQUENTIALLY with Adam (clipnorm 1.0, eps 1e-7) followed by the
    Keras max_norm constraints, and ε decays by ``epsilon_decay`` — the
    reference's ``replay``;
  * checkpoints land in the compatible ``model_ChebConv_*`` layout.

On GPU the per-sample sets come from the fused ChebConv backward's
per-graph partials; on CPU from masked-cotangent backwards (slower,
used for testing and small runs).
"""

from __future__ import annotations

import argparse
import json
import os
import random
import time
from collections import deque

import torch

from ..engine import EpisodeEngine
from ..models.chebconv import ChebConvStack
from ..utils.checkpoint import model_dir
from ..utils import checkpoint as ckpt_io
from .train_batched import build_training_cases


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=2000,
                    help="episode-batch steps (each contributes B gradient "
                         "sets to the replay memory)")
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--sizes", type=str,
                    default="20,30,40,50,60,70,80,90,100,110")
    ap.add_argument("--distinct", type=int, default=64)
    ap.add_argument("--T", type=int, default=1000)
    ap.add_argument("--arrival_scale", type=float, default=0.15)
    ap.add_argument("--learning_rate", type=float, default=1e-6,
                    help="the reference train.sh value; each replay applies\n                    ~replay_batch sequential Adam steps, so keep it small")
    ap.add_argument("--K", type=int, default=2)
    ap.add_argument("--replay_batch", type=int, default=100)
    ap.add_argument("--replay_every", type=int, default=1)
    ap.add_argument("--memory", type=int, default=5000)
    ap.add_argument("--epsilon", type=float, default=0.1)
    ap.add_argument("--epsilon_decay", type=float, default=0.985)
    ap.add_argument("--epsilon_min", type=float, default=0.001)
    ap.add_argument("--seed", type=int, default=100)
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--training_set", type=str, default="BATRPL")
    ap.add_argument("--model_root", type=str, default="model")
    ap.add_argument("--save_every", type=int, default=500)
    ap.add_argument("--log_every", type=int, default=50)
    ap.add_argument("--init_scale", type=float, default=0.01)
    ap.add_argument("--device", type=str, default=None)
    args = ap.parse_args(argv)

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    dtype = torch.float32 if device.startswith("cuda") else torch.float64
    model = ChebConvStack(K=args.K, dtype=dtype, seed=args.seed)
    with torch.no_grad():
        for layer in model.layers:
            layer.weight.mul_(args.init_scale)
        model.layers[-1].bias.fill_(0.5)

    sizes = [int(s) for s in args.sizes.split(",")]
    engines = [EpisodeEngine(
        build_training_cases(n, max(args.batch // len(sizes), 8),
                             args.distinct, args.T,
                             args.seed + 17 * n, workers=args.workers),
        model, device=device, dtype=dtype) for n in sizes]
    opt = torch.optim.Adam(model.parameters(), lr=args.learning_rate,
                           eps=1e-7)
    gen = torch.Generator(device=device)
    gen.manual_seed(args.seed * 977)
    rng = random.Random(args.seed)

    memory = deque(maxlen=args.memory)
    epsilon = args.epsilon
    actor_dir = model_dir(args.model_root, args.training_set)
    t0 = time.time()
    history = []
    episodes = 0

    def replay():
        nonlocal epsilon
        if len(memory) < args.replay_batch:
            return
        sample = rng.sample(list(memory), args.replay_batch)
        with torch.no_grad():
            for gset in sample:
                for p, gr in zip(model.parameters(), gset):
                    n = gr.norm().clamp(min=1e-12)
                    p.grad = gr * (torch.clamp(n, max=1.0) / n)
                opt.step()
                model.apply_constraints()
        if epsilon > args.epsilon_min:
            epsilon *= args.epsilon_decay

    for step in range(1, args.steps + 1):
        eng = engines[step % len(engines)]
        jobs = eng.sample_jobs(args.arrival_scale, gen)
        for p in model.parameters():
            p.grad = None
        res = eng.gnn_episode(jobs, explore=epsilon, gen=gen, train=True,
                              per_sample=True)
        memory.extend(tuple(g.clone() for g in gs)
                      for gs in eng.last_per_sample_grads)
        episodes += eng.B
        if step % args.replay_every == 0:
            replay()
        if step % args.log_every == 0:
            rec = {"step": step, "tau": float(torch.nanmean(res.tau)),
                   "congest_ratio": float(res.congest.sum())
                   / max(float(res.num_jobs.sum()), 1.0),
                   "epsilon": epsilon, "memory": len(memory),
                   "eps_per_sec": episodes / (time.time() - t0)}
            history.append(rec)
            print(json.dumps(rec), flush=True)
        if step % args.save_every == 0:
            ckpt_io.save(model, os.path.join(
                actor_dir, f"cp-{step // args.save_every:04d}.ckpt"))
    ckpt_io.save(model, os.path.join(actor_dir, "cp-9999.ckpt"))
    return history


if __name__ == "__main__":
    main()
