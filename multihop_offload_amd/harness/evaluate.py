"""Batched evaluator: reproduce the reference's headline table
(BASELINE.md / SURVEY.md §6) — mean task latency τ and task congestion
ratio for baseline / local / GNN over 20–110-node BA cases at a given load.

Uses the device-resident engine per node-size bucket (each bucket is a
same-N batch); GNN runs forward-only with the loaded checkpoint.

Run:  python -m multihop_offload_amd.harness.evaluate \
          --training_set BAT1000 --instances 10 --cases-per-size 20
"""

from __future__ import annotations

import argparse
import json
import os

import torch

from ..engine import EpisodeEngine
from ..models.chebconv import ChebConvStack
from ..utils.checkpoint import latest_checkpoint, load, model_dir
from .train_batched import build_training_cases


def evaluate(model, sizes, cases_per_size, instances, T, load, seed,
             device, dtype, workers: int = 8, lam_margin: float = 0.0,
             refine: int = 0):
    """Returns per-method aggregate {tau, congest_jobs, num_jobs} summed /
    averaged over all (size, case, instance)."""
    agg = {m: {"tau_sum": 0.0, "tau_n": 0, "congest": 0, "jobs": 0,
               "ratio_sum": 0.0, "ratio_n": 0}
           for m in ("baseline", "local", "GNN")}
    per_size = {}
    for n in sizes:
        cases = build_training_cases(n, cases_per_size, cases_per_size, T,
                                     seed + n, workers=workers)
        engine = EpisodeEngine(cases, model, device=device, dtype=dtype,
                               lam_margin=lam_margin)
        gen = torch.Generator(device=device)
        gen.manual_seed(seed + n)
        ps = {m: {"tau_sum": 0.0, "tau_n": 0, "congest": 0, "jobs": 0,
                  "ratio_sum": 0.0, "ratio_n": 0}
              for m in ("baseline", "local", "GNN")}
        for _ in range(instances):
            jobs = engine.sample_jobs(load, gen)
            results = {
                "baseline": engine.baseline_episode(jobs),
                "local": engine.local_episode(jobs),
                "GNN": engine.gnn_episode(jobs, train=False,
                                          refine=refine),
            }
            bl = results["baseline"].delay_emp
            for m, res in results.items():
                # per-task latency ratio vs baseline (notebook cell 16)
                r = res.delay_emp / bl
                ok = torch.isfinite(r)
                for d in (agg[m], ps[m]):
                    d["tau_sum"] += float(torch.nansum(res.tau))
                    d["tau_n"] += int(torch.isfinite(res.tau).sum())
                    d["congest"] += int(res.congest.sum())
                    d["jobs"] += int(res.num_jobs.sum())
                    d["ratio_sum"] += float(r[ok].sum())
                    d["ratio_n"] += int(ok.sum())
        engine.check_overflow()       # strict: truncated walks void an eval
        per_size[n] = {
            m: {"tau": d["tau_sum"] / max(d["tau_n"], 1),
                "congest_ratio": d["congest"] / max(d["jobs"], 1),
                "latency_ratio": d["ratio_sum"] / max(d["ratio_n"], 1)}
            for m, d in ps.items()}
    summary = {
        m: {"tau": d["tau_sum"] / max(d["tau_n"], 1),
            "congest_ratio": d["congest"] / max(d["jobs"], 1),
            "latency_ratio": d["ratio_sum"] / max(d["ratio_n"], 1),
            "jobs": d["jobs"]}
        for m, d in agg.items()}
    return summary, per_size


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--training_set", type=str, default="BAT1000")
    ap.add_argument("--model_root", type=str, default="model")
    ap.add_argument("--sizes", type=str, default="20,30,40,50,60,70,80,90,100,110")
    ap.add_argument("--cases-per-size", type=int, default=20)
    ap.add_argument("--instances", type=int, default=10)
    ap.add_argument("--T", type=int, default=1000)
    ap.add_argument("--load", type=float, default=0.15)
    ap.add_argument("--K", type=int, default=2)
    ap.add_argument("--seed", type=int, default=500)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--lam_margin", type=float, default=0.0,
                    help="inference-time conservatism: inflate predicted "
                         "traffic by (1+margin) at decision time "
                         "(congestion-tail control; 0 = reference)")
    ap.add_argument("--refine", type=int, default=0,
                    help="congestion-aware refinement passes: jobs whose "
                         "analytic delay exceeds T fall back to local and "
                         "the assignment is re-evaluated (NOT reference "
                         "semantics; reported results must say so)")
    ap.add_argument("--out", type=str, default="out/eval_summary.json")
    args = ap.parse_args(argv)

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    dtype = torch.float32 if device.startswith("cuda") else torch.float64
    model = ChebConvStack(K=args.K, dtype=dtype)
    ckpt = latest_checkpoint(model_dir(args.model_root, args.training_set))
    if ckpt:
        load(model, ckpt)
        print(f"loaded {ckpt}")
    else:
        print("WARNING: no checkpoint found — evaluating random init")

    sizes = [int(s) for s in args.sizes.split(",")]
    summary, per_size = evaluate(model, sizes, args.cases_per_size,
                                 args.instances, args.T, args.load,
                                 args.seed, device, dtype, args.workers,
                                 lam_margin=args.lam_margin,
                                 refine=args.refine)
    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    with open(args.out, "w") as f:
        json.dump({"summary": summary, "per_size": per_size,
                   "config": vars(args)}, f, indent=1)
    print(json.dumps(summary, indent=1))
    return summary


if __name__ == "__main__":
    main()
