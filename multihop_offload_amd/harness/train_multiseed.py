"""Multi-seed production trainer.

The semi-analytic policy gradient is bistable across seeds (docs/TRAINING.md
"Stability outcomes": 3/10 seeds reach the flagship-grade region, the rest
are guard-contained at conservative grade).  The production recipe is
therefore a small seed population: run `--seeds` independent
`train_batched` runs (~3 min each on one MI355X at the flagship recipe),
evaluate every candidate's best-eval checkpoint on ONE shared fresh
held-out set, and ship the winner into `--training_set`.

Run:  python -m multihop_offload_amd.harness.train_multiseed \
          --seeds 210,230,250 --steps 40000 --batch 1536 \
          --sizes 20,30,40,50,60,70,80,90,100,110,120 --pad_mixed
"""

from __future__ import annotations

import argparse
import json
import os
import shutil

import torch


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--seeds", type=str, default="210,230,250")
    ap.add_argument("--training_set", type=str, default="BEST",
                    help="the winner ships as model_ChebConv_<this>_*")
    ap.add_argument("--model_root", type=str, default="model")
    ap.add_argument("--select_cases", type=int, default=24,
                    help="fresh held-out cases per size for the final "
                         "cross-seed selection")
    ap.add_argument("--select_instances", type=int, default=6)
    ap.add_argument("--select_seed", type=int, default=424242,
                    help="seed of the SHARED selection set (identical for "
                         "every candidate)")
    ap.add_argument("--congest_weight", type=float, default=3000.0)
    args, rest = ap.parse_known_args(argv)

    from . import train_batched
    from .evaluate import evaluate
    from ..models.chebconv import ChebConvStack
    from ..utils.checkpoint import latest_checkpoint, load, model_dir

    # parse shared trainer args once to learn the eval geometry
    tparser_probe = argparse.ArgumentParser()
    tparser_probe.add_argument("--sizes", type=str, default=None)
    tparser_probe.add_argument("--nodes", type=int, default=100)
    tparser_probe.add_argument("--T", type=int, default=1000)
    tparser_probe.add_argument("--arrival_scale", type=float, default=0.15)
    tparser_probe.add_argument("--K", type=int, default=2)
    probe, _ = tparser_probe.parse_known_args(rest)
    sizes = ([int(s) for s in probe.sizes.split(",")] if probe.sizes
             else [probe.nodes])

    seeds = [int(s) for s in args.seeds.split(",")]
    results = []
    for seed in seeds:
        name = f"{args.training_set}S{seed}"
        train_batched.main(rest + [
            "--seed", str(seed), "--training_set", name,
            "--model_root", args.model_root])
        # evaluate the candidate's best-eval checkpoint on the SHARED set
        device = "cuda" if torch.cuda.is_available() else "cpu"
        dtype = (torch.float32 if device == "cuda" else torch.float64)
        model = ChebConvStack(K=probe.K, dtype=dtype)
        ckpt = latest_checkpoint(model_dir(args.model_root, name))
        load(model, ckpt)
        # exclude sizes > 110 from selection (the eval convention)
        esizes = [n for n in sizes if n <= 110] or sizes
        summary, _ = evaluate(model, esizes, args.select_cases,
                              args.select_instances, probe.T,
                              probe.arrival_scale, args.select_seed,
                              device, dtype)
        g = summary["GNN"]
        metric = g["tau"] + args.congest_weight * g["congest_ratio"]
        rec = {"seed": seed, "name": name, "ckpt": ckpt,
               "tau": g["tau"], "congest_ratio": g["congest_ratio"],
               "metric": metric}
        results.append(rec)
        print(json.dumps(rec), flush=True)

    best = min(results, key=lambda r: r["metric"])
    dst = model_dir(args.model_root, args.training_set)
    os.makedirs(dst, exist_ok=True)
    src_dir = os.path.dirname(best["ckpt"])
    for f in os.listdir(src_dir):
        shutil.copy2(os.path.join(src_dir, f), os.path.join(dst, f))
    with open(os.path.join(args.model_root,
                           f"multiseed_{args.training_set}.json"),
              "w") as f:
        json.dump({"results": results, "winner": best}, f, indent=1)
    print(json.dumps({"winner": best}), flush=True)
    return results


if __name__ == "__main__":
    main()
