"""MI355X-native batched trainer.

The reference trains one instance at a time (~0.25 s each,
``AdHoc_train.py``); this trainer drives the device-resident engine: every
step processes a batch of (graph, jobs) episodes, computes the summed
semi-analytic actor gradient in one fused backward, all-reduces it across
data-parallel ranks (RCCL/xGMI) and applies Adam with the reference's
constraint semantics.  Checkpoints land in the compatible
``model_ChebConv_*`` layout.

Run:  python -m multihop_offload_amd.harness.train_batched --steps 2000
      (multi-GPU: python -m torch.distributed.run --nproc-per-node 8 ...)
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch

from ..engine import EpisodeEngine
from ..models.chebconv import ChebConvStack
from ..parallel import dp
from ..utils.checkpoint import model_dir
from ..utils import checkpoint as ckpt_io


def _gen_one(task):
    n_nodes, seed, gtype = task
    import numpy as np
    from ..datagen import generate_case
    return generate_case(n_nodes, seed, gtype,
                         rng=np.random.RandomState(seed))[0]


def build_training_cases(n_nodes, batch, distinct, T, seed, gtype="ba",
                         workers: int = 0):
    """`distinct` full topologies (role assignment per datagen
    distributions, optionally generated in parallel), replicated to `batch`
    cases with independent link-rate draws (structure shared)."""
    from ..graphs import CaseGraph
    if distinct > batch:
        # bases beyond the batch size would never be instantiated — cap
        # (a batch of B cases can carry at most B distinct topologies)
        distinct = batch
    rng = np.random.RandomState(seed)
    tasks = [(n_nodes, seed + d, gtype) for d in range(distinct)]
    if workers > 1:
        import multiprocessing as mp
        with mp.get_context("fork").Pool(workers) as pool:
            protos = pool.map(_gen_one, tasks)
    else:
        protos = [_gen_one(t) for t in tasks]

    bases = []
    for d, case in enumerate(protos):
        adj = np.asarray(case["adj"].todense())
        g = CaseGraph(n_nodes, t_max=T, seed=seed + d, gtype=gtype,
                      adj=adj, pos=case["pos_c"])
        for nidx in range(n_nodes):
            role, bw = case["nodes_info"][nidx, 0], float(
                case["nodes_info"][nidx, 1])
            if role == 2:
                g.add_relay(nidx)
            elif role == 1:
                g.add_server(nidx, bw)
            else:
                g.set_mobile_bw(nidx, bw)
        g.links_init(case["link_rate"], rng=rng)
        g.sp_hop        # materialize before cloning so replicas share it
        g.ext
        bases.append((g, case["link_rate"]))

    cases = []
    for b in range(batch):
        g, base_rates = bases[b % distinct]
        cases.append(g if b < distinct
                     else g.clone_with_rates(base_rates, rng))
    return cases


def _dist_mean(x: float, world: int, device) -> float:
    """Average a scalar across data-parallel ranks so control-flow
    decisions (divergence guard, eval-based selection) are identical on
    every rank — per-rank decisions would silently de-synchronise the
    replicated parameters."""
    if world > 1 and torch.distributed.is_initialized():
        dev = device if torch.distributed.get_backend() == "nccl" else "cpu"
        t = torch.tensor([x], dtype=torch.float32, device=dev)
        torch.distributed.all_reduce(t)
        return float(t.item()) / world
    return float(x)


def evaluate_policy(engines, loads, seed: int, rounds: int = 4):
    """Held-out evaluation: mean per-job tau AND congestion ratio of the
    greedy policy (explore=0, no gradients) over fresh job draws from a
    FIXED seed, so successive calls during training are comparable."""
    taus = []
    congest = jobs_n = 0
    with torch.no_grad():
        for engine in engines:
            gen = torch.Generator(device=engine.device)
            gen.manual_seed(seed)
            for r in range(rounds):
                jobs = engine.sample_jobs(loads[r % len(loads)], gen)
                res = engine.gnn_episode(jobs, train=False)
                taus.append(res.tau.flatten())
                congest += int(res.congest.sum())
                jobs_n += int(res.num_jobs.sum())
    return (float(torch.nanmean(torch.cat(taus))),
            congest / max(jobs_n, 1))


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=2000)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--nodes", type=int, default=100)
    ap.add_argument("--sizes", type=str, default=None,
                    help="comma-separated node counts for mixed-size "
                         "training (one engine per size, round-robin; "
                         "overrides --nodes)")
    ap.add_argument("--distinct", type=int, default=32)
    ap.add_argument("--T", type=int, default=1000)
    ap.add_argument("--arrival_scale", type=float, default=0.15)
    ap.add_argument("--arrival_scales", type=str, default=None,
                    help="comma-separated loads cycled per step (e.g. "
                         "0.15,0.15,0.15,0.20) — overrides --arrival_scale")
    ap.add_argument("--learning_rate", type=float, default=1e-4)
    ap.add_argument("--K", type=int, default=2)
    ap.add_argument("--explore", type=float, default=0.1)
    ap.add_argument("--explore_decay", type=float, default=0.999)
    ap.add_argument("--seed", type=int, default=100)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--dtype", type=str, default=None,
                    choices=[None, "float32", "float64"],
                    help="override the compute dtype (default: fp32 on "
                         "GPU, fp64 on CPU).  CPU fp32 vs fp64 with "
                         "identical seeds is the precision drift study "
                         "(docs/KERNELS.md)")
    ap.add_argument("--training_set", type=str, default="BAT1000")
    ap.add_argument("--model_root", type=str, default="model")
    ap.add_argument("--save_every", type=int, default=500)
    ap.add_argument("--log_every", type=int, default=50)
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--lr_decay_at", type=str, default="0",
                    help="step(s) after which lr is multiplied by 0.1 "
                         "(comma-separated for multi-decay schedules, "
                         "e.g. 30000,70000; 0 disables)")
    ap.add_argument("--guard_every", type=int, default=1000,
                    help="divergence guard period (0 disables): if the "
                         "running tau explodes past 5x the best seen, roll "
                         "back to the best parameters and cut lr 3x")
    ap.add_argument("--init_scale", type=float, default=0.01,
                    help="shrink initial weights (wakes the output ReLU)")
    ap.add_argument("--pad_mixed", action="store_true",
                    help="mixed --sizes in ONE engine (smaller cases "
                         "padded with inert relay nodes to the max size): "
                         "every step trains on the full mixed batch "
                         "instead of round-robining per-size engines")
    ap.add_argument("--eval_every", type=int, default=0,
                    help="held-out greedy evaluation period (0 disables); "
                         "the final checkpoint is then the best-by-eval-tau "
                         "parameters, a stronger selection signal than the "
                         "training tau the guard tracks")
    ap.add_argument("--eval_rounds", type=int, default=4)
    ap.add_argument("--eval_cases", type=int, default=0,
                    help="held-out eval cases per size (0 = min(per-size "
                         "batch, 16)); larger sets give a less noisy "
                         "selection signal")
    ap.add_argument("--eval_on_train", action="store_true",
                    help="evaluate on the training topologies instead of "
                         "the default held-out ones (held-out catches "
                         "topology overfitting — see docs/TRAINING.md)")
    ap.add_argument("--eval_seed", type=int, default=12345)
    ap.add_argument("--eval_guard", type=float, default=3.0,
                    help="held-out divergence guard: if the eval metric "
                         "exceeds this multiple of the best seen, roll "
                         "back to the best-eval parameters and cut lr 3x "
                         "(0 disables).  Catches the bistable collapse "
                         "that stays invisible on TRAINING topologies "
                         "(train tau ~17 while held-out tau >100)")
    ap.add_argument("--eval_congest_weight", type=float, default=3000.0,
                    help="held-out selection metric = eval tau + this "
                         "weight x eval congestion ratio (the north-star "
                         "quality clause is the congestion TAIL, which "
                         "tau alone does not rank); 0 reverts to pure tau")
    ap.add_argument("--torch_profile", type=int, default=0,
                    help="profile N steps with torch.profiler after warmup "
                         "and write a chrome trace next to the model dir")
    ap.add_argument("--resume", action="store_true",
                    help="load the latest checkpoint from the model dir "
                         "before training (reference resume protocol)")
    ap.add_argument("--delay_clamp", type=float, default=0.0,
                    help="clamp the 1/(mu-lam) delay branch of the "
                         "DIFFERENTIABLE path at this value (0 = off): "
                         "pole mitigation for long-horizon stability "
                         "(docs/TRAINING.md); the empirical evaluator is "
                         "never clamped")
    ap.add_argument("--capture", action="store_true",
                    help="hipGraph-capture the whole training step "
                         "(sample→episode→all-reduce→fused Adam) and "
                         "replay it; explore lives in a device scalar so "
                         "its decay survives capture; graphs are "
                         "re-captured on lr changes")
    args = ap.parse_args(argv)

    rank, world = dp.init_from_env()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    dtype = (getattr(torch, args.dtype) if args.dtype
             else (torch.float32 if device.startswith("cuda")
                   else torch.float64))

    model = ChebConvStack(K=args.K, dtype=dtype, seed=args.seed)
    resumed = False
    if args.resume:
        from ..utils.checkpoint import latest_checkpoint
        latest = latest_checkpoint(model_dir(args.model_root,
                                             args.training_set))
        if latest:
            ckpt_io.load(model, latest)
            resumed = True
            if rank == 0:
                print(f"resumed from {latest}", flush=True)
        elif rank == 0:
            print("no checkpoint to resume — fresh init", flush=True)
    if args.init_scale != 1.0 and not resumed:
        with torch.no_grad():
            for layer in model.layers:
                layer.weight.mul_(args.init_scale)
            model.layers[-1].bias.fill_(0.5)

    sizes = ([int(s) for s in args.sizes.split(",")] if args.sizes
             else [args.nodes])
    per_size = max(args.batch // len(sizes), 8)
    if args.pad_mixed and len(sizes) > 1:
        n_max = max(sizes)
        cases = []
        for n in sizes:
            cases += [c.pad_to(n_max) for c in build_training_cases(
                n, per_size, args.distinct, args.T,
                args.seed + 1000 * rank + 17 * n, workers=args.workers)]
        engines = [EpisodeEngine(cases, model, device=device, dtype=dtype,
                                 delay_clamp=args.delay_clamp)]
    else:
        engines = []
        for n in sizes:
            cases = build_training_cases(
                n, per_size, args.distinct, args.T,
                args.seed + 1000 * rank + 17 * n, workers=args.workers)
            engines.append(EpisodeEngine(cases, model, device=device,
                                         dtype=dtype,
                                         delay_clamp=args.delay_clamp))
    engine = engines[0]
    dp.broadcast_params(engine.model)
    from ..ops import dispatch as mho_dispatch
    use_fused = (device.startswith("cuda") and mho_dispatch.hip_available()
                 and dtype == torch.float32)
    if use_fused:
        # one kernel: grad scale -> clipnorm -> Adam -> max_norm; flat_g is
        # the single RCCL payload (same path bench.py measures)
        from ..ops.functions import FusedAdam
        opt = FusedAdam(engine.model, lr=args.learning_rate)
        reducer = None
    else:
        opt = torch.optim.Adam(engine.model.parameters(),
                               lr=args.learning_rate, eps=1e-7)
        reducer = dp.FlatAllreduce(engine.model.parameters())
    gen = torch.Generator(device=device)
    gen.manual_seed(args.seed * 7919 + rank)
    for i, e in enumerate(engines):
        e.set_rng_seed(args.seed * 7919 + 131 * rank + i)

    eval_engines = engines
    if args.eval_every and not args.eval_on_train:
        # held-out topologies (unseen seeds): training-set eval hides
        # topology overfitting (docs/TRAINING.md, LONGSOAK exhibit)
        ec = args.eval_cases or min(per_size, 16)
        ed = args.eval_cases or min(args.distinct, 16)
        eval_engines = [EpisodeEngine(
            build_training_cases(n, ec, ed, args.T,
                                 args.seed + 900000 + 17 * n,
                                 workers=args.workers),
            model, device=device, dtype=dtype) for n in sizes]

    lr_decay_steps = {int(x) for x in str(args.lr_decay_at).split(",")
                      if int(x or 0) > 0}
    actor_dir = model_dir(args.model_root, args.training_set)
    explore = args.explore
    t0 = time.time()
    history = []
    # divergence guard state (the semi-analytic gradient is unstable near
    # the 1/(mu-lambda) poles; see ROUND2.md — rollback + lr cut recovers)
    import copy as _copy
    best_tau = float("inf")
    best_params = None
    best_opt = None
    rollbacks = 0
    best_eval_tau = float("inf")
    best_eval_params = None
    best_eval_opt = None
    eval_rollbacks = 0
    eval_hist = []
    last_rollback_step = -10**9
    loads = ([float(x) for x in args.arrival_scales.split(",")]
             if args.arrival_scales else [args.arrival_scale])
    profiler = None
    profiler_on = False
    if args.torch_profile and rank == 0:
        from torch.profiler import profile, ProfilerActivity
        acts = [ProfilerActivity.CPU]
        if device.startswith("cuda"):
            acts.append(ProfilerActivity.CUDA)
        profiler = profile(activities=acts, acc_events=True)

    # ---- hipGraph capture of the whole training step ----------------------
    # explore lives in a device scalar (the captured kernels read its
    # CURRENT value), randomness comes from torch's graph-safe default
    # generator plus the engine's counter RNG, and lr changes invalidate
    # the captured graphs (the fused-Adam lr is baked at capture time).
    use_capture = (args.capture and use_fused
                   and device.startswith("cuda"))
    explore_dev = None
    if use_capture:
        assert world == 1 or True  # all-reduce capture is supported by RCCL
        explore_dev = torch.tensor(float(args.explore),
                                   dtype=torch.float32, device=device)
        torch.manual_seed(args.seed * 7919 + rank)
        gen = None
    hip_graphs = {}

    def _fused_step(eng, load, explore_arg):
        opt.zero_grad()
        jobs = eng.sample_jobs(load, gen)
        res = eng.gnn_episode(jobs, explore=explore_arg, gen=gen, train=True)
        if world > 1:
            torch.distributed.all_reduce(opt.flat_g)
        opt.step(scale=1.0 / (eng.B * world))
        return res

    def _captured_step(eng_idx, load_idx):
        key = (eng_idx, load_idx)
        if key not in hip_graphs:
            eng = engines[eng_idx]
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            _fused_step(eng, loads[load_idx], explore_dev)  # warm side effects
            with torch.cuda.graph(g):
                res = _fused_step(eng, loads[load_idx], explore_dev)
            hip_graphs[key] = (g, res)
        g, res = hip_graphs[key]
        g.replay()
        return res

    for step in range(1, args.steps + 1):
        if profiler is not None and not profiler_on and \
                step >= max(args.steps - args.torch_profile, 1):
            profiler.__enter__()
            profiler_on = True
        engine = engines[step % len(engines)]
        if use_capture:
            res = _captured_step(step % len(engines), step % len(loads))
            explore_dev.mul_(args.explore_decay).clamp_(min=0.001)
            pre_norms = None
        elif use_fused:
            jobs = engine.sample_jobs(loads[step % len(loads)], gen)
            opt.zero_grad()
            res = engine.gnn_episode(jobs, explore=explore, gen=gen,
                                     train=True)
            scale = 1.0 / (engine.B * world)
            if world > 1:
                torch.distributed.all_reduce(opt.flat_g)
            with torch.no_grad():
                pre_norms = [float(p.grad.norm()) * scale
                             for p in engine.model.parameters()]
            opt.step(scale=scale)
        else:
            jobs = engine.sample_jobs(loads[step % len(loads)], gen)
            for p in engine.model.parameters():
                p.grad = None
            res = engine.gnn_episode(jobs, explore=explore, gen=gen,
                                     train=True)
            with torch.no_grad():
                for p in engine.model.parameters():
                    if p.grad is not None:
                        p.grad /= engine.B
            reducer(average=True)
            with torch.no_grad():
                pre_norms = [float(p.grad.norm())
                             for p in engine.model.parameters()
                             if p.grad is not None]
                for p in engine.model.parameters():
                    if p.grad is not None:
                        n = p.grad.norm().clamp(min=1e-12)
                        if not torch.isfinite(n):
                            p.grad.zero_()   # poisoned step: skip tensor
                            continue
                        p.grad *= torch.clamp(n, max=1.0) / n
            opt.step()
            engine.model.apply_constraints()
        explore = max(explore * args.explore_decay, 0.001)
        if step in lr_decay_steps:
            for group in opt.param_groups:
                group["lr"] *= 0.1
            hip_graphs.clear()      # lr is baked into captured graphs
        if args.eval_every and step % args.eval_every == 0:
            et, ec = evaluate_policy(eval_engines, loads, args.eval_seed,
                                     args.eval_rounds)
            eval_tau = _dist_mean(et, world, engine.device)
            eval_congest = _dist_mean(ec, world, engine.device)
            eval_metric = eval_tau + args.eval_congest_weight * eval_congest
            rolled = 0
            # dead-policy rescue: a collapsed run can freeze in a
            # zero-output attractor (dead output ReLU → zero gradients →
            # bit-identical eval forever, observed at seed-dependent
            # rates).  Three identical evals ⇒ re-wake the output bias
            # and re-explore.
            eval_hist.append(eval_metric)
            dead = (len(eval_hist) >= 3
                    and max(eval_hist[-3:]) - min(eval_hist[-3:]) < 1e-9)
            if eval_metric < best_eval_tau:
                best_eval_tau = eval_metric
                best_eval_params = [p.detach().clone()
                                    for p in engine.model.parameters()]
                best_eval_opt = _copy.deepcopy(opt.state_dict())
            elif dead or (args.eval_guard and best_eval_params is not None
                          and eval_metric > args.eval_guard
                          * max(best_eval_tau, 20.0)):
                # held-out collapse: restore the best-eval state (keep the
                # SCHEDULE lr — restoring the snapshot's lr would undo
                # decays), re-explore, and cool the lr only when collapses
                # repeat back-to-back (a single excursion self-recovers;
                # compounding cuts freeze the run at its early best)
                cur_lr = opt.param_groups[0]["lr"]
                if best_eval_params is not None:
                    with torch.no_grad():
                        for p, bp in zip(engine.model.parameters(),
                                         best_eval_params):
                            p.copy_(bp)
                    if best_eval_opt is not None:
                        opt.load_state_dict(best_eval_opt)
                if dead:
                    # frozen policy (possibly the restored best itself, if
                    # the run died at init): re-wake the output layer
                    with torch.no_grad():
                        engine.model.layers[-1].bias.fill_(0.5)
                        if best_eval_params is not None and \
                                best_eval_tau >= 200.0:
                            # the "best" is the dead policy — drop it so a
                            # woken policy can take over
                            best_eval_tau = float("inf")
                            best_eval_params = None
                            best_eval_opt = None
                repeated = (step - last_rollback_step
                            <= 3 * args.eval_every)
                for group in opt.param_groups:
                    group["lr"] = cur_lr / (3.0 if repeated else 1.0)
                explore = max(explore, 0.05)
                if explore_dev is not None:
                    explore_dev.fill_(max(float(explore_dev), 0.05))
                last_rollback_step = step
                eval_hist.clear()
                hip_graphs.clear()
                eval_rollbacks += 1
                rolled = 1
            rec = {"step": step, "eval_tau": eval_tau,
                   "eval_congest": eval_congest,
                   "eval_metric": eval_metric,
                   "best_eval_metric": best_eval_tau,
                   "eval_rollbacks": eval_rollbacks,
                   "lr": opt.param_groups[0]["lr"],
                   "rolled": rolled}
            history.append(rec)        # every rank: values are all-reduced
            if rank == 0:
                print(json.dumps(rec), flush=True)
        if args.guard_every and step % args.guard_every == 0:
            tau_now = _dist_mean(float(torch.nanmean(res.tau)),
                                 world, engine.device)
            if tau_now < best_tau:
                best_tau = tau_now
                best_params = [p.detach().clone()
                               for p in engine.model.parameters()]
                best_opt = _copy.deepcopy(opt.state_dict())
            elif best_params is not None and \
                    tau_now > 5.0 * max(best_tau, 20.0):
                with torch.no_grad():
                    for p, bp in zip(engine.model.parameters(), best_params):
                        p.copy_(bp)
                opt.load_state_dict(best_opt)
                for group in opt.param_groups:
                    group["lr"] /= 3.0
                hip_graphs.clear()  # lr is baked into captured graphs
                rollbacks += 1
                if rank == 0:
                    print(json.dumps({"step": step, "rollback": rollbacks,
                                      "tau_now": tau_now,
                                      "best_tau": best_tau,
                                      "lr": opt.param_groups[0]["lr"]}),
                          flush=True)

        if step % args.log_every == 0 and rank == 0:
            if pre_norms is None:   # capture mode: flat_g still holds the
                # replayed step's summed gradients
                with torch.no_grad():
                    pre_norms = [float(p.grad.norm()) / (engine.B * world)
                                 for p in engine.model.parameters()]
            tau = float(torch.nanmean(res.tau))
            congest = int(res.congest.sum())
            njobs = int(res.num_jobs.sum())
            if use_fused:
                # clipping happens inside the kernel: reconstruct the
                # post-clip global norm from the scaled pre-clip norms
                gnorm = float(sum(min(n, 1.0) ** 2
                                  for n in pre_norms) ** 0.5)
            else:
                gnorm = float(sum((p.grad ** 2).sum()
                              for p in engine.model.parameters()
                              if p.grad is not None) ** 0.5)
            # pre-clip norms are the stability signal: the post-clip norm
            # is pinned at sqrt(#tensors) whenever every tensor saturates
            # the per-tensor clip (see ROUND2.md stability probe)
            rec = {"step": step, "tau": tau,
                   "walk_overflow": sum(e.check_overflow(strict=False)
                                        for e in engines),
                   "grad_norm_preclip_max": max(pre_norms) if pre_norms
                   else 0.0,
                   "grad_norm_preclip_sum": float(np.hypot.reduce(
                       pre_norms)) if pre_norms else 0.0,
                   "congest_ratio": congest / max(njobs, 1),
                   "loss_fn": float(res.loss_fn), "loss_mse": float(res.loss_mse),
                   "grad_norm": gnorm,
                   "explore": explore,
                   "eps_per_sec": engine.B * world * step
                   / (time.time() - t0)}
            history.append(rec)
            print(json.dumps(rec), flush=True)
        if step % args.save_every == 0 and rank == 0:
            ckpt_io.save(engine.model,
                         os.path.join(actor_dir,
                                      f"cp-{step // args.save_every:04d}.ckpt"))
    if profiler_on:
        profiler.__exit__(None, None, None)
        trace = os.path.join(args.model_root,
                             f"torch_trace_{args.training_set}.json")
        profiler.export_chrome_trace(trace)
        print(f"torch.profiler trace -> {trace}", flush=True)
    if rank == 0:
        # ship the best-seen parameters: held-out-eval selection when
        # --eval_every was on, else the guard's best-training-tau snapshot
        ship = best_eval_params if best_eval_params is not None else best_params
        if ship is not None:
            with torch.no_grad():
                for p, bp in zip(engine.model.parameters(), ship):
                    p.copy_(bp)
        ckpt_io.save(engine.model, os.path.join(actor_dir, "cp-9999.ckpt"))
        with open(os.path.join(args.model_root,
                               f"train_history_{args.training_set}.json"),
                  "w") as f:
            json.dump(history, f, indent=1)
    return history


if __name__ == "__main__":
    main()
