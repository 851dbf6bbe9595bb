#!/usr/bin/env python3
"""Per-stage kernel micro-benchmark (GPU): times each fused kernel of the
episode step in isolation with hip events, at the flagship config
(N=110 BA, B=1024) and the large-graph config (N=1000 ER, B=64).

Run on an MI355X:  python profiles/bench_kernels.py [--config flagship|er1000]
Output: one JSON line per (config, stage) with mean µs over the timed reps.
"""
import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timeit(fn, reps=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) * 1000.0 / reps   # µs


def bench_config(name, nodes, batch, gtype, distinct, reps):
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.harness.train_batched import build_training_cases

    cases = build_training_cases(nodes, batch, distinct, 1000, 7,
                                 gtype=gtype, workers=8)
    model = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    with torch.no_grad():
        for p in model.parameters():
            p.mul_(0.01)
        model.layers[-1].bias.fill_(0.5)
    eng = EpisodeEngine(cases, model, device="cuda", dtype=torch.float32)
    gen = torch.Generator(device="cuda")
    gen.manual_seed(0)
    jobs = eng.sample_jobs(0.15, gen)

    out = {}
    # full step pieces
    with torch.enable_grad():
        dm, *_ = eng.actor_forward(jobs)
    sp = eng.apsp(dm)
    uds = torch.diagonal(dm.detach(), dim1=1, dim2=2)
    dst, _ = eng.offload_decide(jobs, sp, uds)
    rl, nhop, delay_emp, unit_mtx, written = eng._episode_eval(jobs, dst, sp)

    from multihop_offload_amd.ops.functions import ChebStackFn
    params = []
    for layer in eng.model.layers:
        params += [layer.weight, layer.bias]
    arr = torch.zeros(eng.B, eng.N, dtype=eng.dtype, device=eng.device)
    arr = arr.scatter_add(1, jobs.sources, jobs.rates * jobs.ul)
    f_job = torch.zeros(eng.B, eng.Ee, dtype=eng.dtype, device=eng.device)
    vidx = torch.where(eng.comp_mask, eng.node_vedge,
                       torch.zeros_like(eng.node_vedge))
    f_job = f_job.scatter_add(
        1, vidx, torch.where(eng.comp_mask, arr, torch.zeros_like(arr)))
    x = torch.stack([eng.f_self_loop, eng.f_rate, f_job,
                     eng.f_as_server], dim=-1).contiguous()

    from multihop_offload_amd.ops import dispatch
    ext = dispatch.require_hip()
    Wp = x.new_zeros(5, 2, 32, 32)
    bp = x.new_zeros(5, 32)
    for l in range(5):
        w, b = params[2 * l], params[2 * l + 1]
        Wp[l, :w.shape[0], :w.shape[1], :w.shape[2]] = w
        bp[l, :b.shape[0]] = b
    from multihop_offload_amd.ops.functions import cheb_lds_fits
    small = cheb_lds_fits(eng.Ee, 2)
    if small:
        out["cheb_fwd"] = timeit(lambda: ext.cheb_fwd(
            x, Wp, bp, eng.k_ext_indptr, eng.k_ext_base, eng.k_ext_cols,
            eng.k_ext_max_nnz), reps)
        lam, acts, t1s = ext.cheb_fwd(x, Wp, bp, eng.k_ext_indptr,
                                      eng.k_ext_base, eng.k_ext_cols,
                                      eng.k_ext_max_nnz)
        dlam = torch.randn_like(lam)
        out["cheb_bwd"] = timeit(lambda: ext.cheb_bwd(
            dlam, acts, t1s, Wp, eng.k_ext_indptr, eng.k_ext_base,
            eng.k_ext_cols, eng.k_ext_max_nnz), reps)
    else:
        out["cheb_large_fwd"] = timeit(lambda: ext.cheb_large_fwd(
            x, Wp, bp, eng.k_ext_indptr, eng.k_ext_base, eng.k_ext_cols),
            reps)
        lam, acts = ext.cheb_large_fwd(x, Wp, bp, eng.k_ext_indptr,
                                       eng.k_ext_base, eng.k_ext_cols)
        dlam = torch.randn_like(lam)
        out["cheb_large_bwd"] = timeit(lambda: ext.cheb_large_bwd(
            dlam, acts, Wp, eng.k_ext_indptr, eng.k_ext_base,
            eng.k_ext_cols), reps)

    lam_ext = lam.detach()
    out["actor_head_fwd"] = timeit(lambda: ext.actor_head_fwd(
        lam_ext, eng.k_conf_indptr, eng.k_conf_base, eng.k_conf_cols,
        eng.link_rates.contiguous(), eng.bw_comp.contiguous(), eng.k_edges,
        eng.node_vedge, eng.T_arr.contiguous(), eng.k_E_arr, eng.N,
        eng.fp_iters, 0.0), reps)
    dmk, mu_hist = ext.actor_head_fwd(
        lam_ext, eng.k_conf_indptr, eng.k_conf_base, eng.k_conf_cols,
        eng.link_rates.contiguous(), eng.bw_comp.contiguous(), eng.k_edges,
        eng.node_vedge, eng.T_arr.contiguous(), eng.k_E_arr, eng.N,
        eng.fp_iters, 0.0)
    gd = torch.randn_like(dmk)
    out["actor_head_bwd"] = timeit(lambda: ext.actor_head_bwd(
        gd, lam_ext, mu_hist, eng.k_conf_indptr, eng.k_conf_base,
        eng.k_conf_cols, eng.link_rates.contiguous(),
        eng.bw_comp.contiguous(), eng.k_edges, eng.node_vedge,
        eng.T_arr.contiguous(), eng.k_E_arr, eng.fp_iters, 0.0), reps)

    out["fw_apsp"] = timeit(lambda: eng.apsp(dm.detach()), reps)
    out["decide"] = timeit(lambda: eng.offload_decide(jobs, sp, uds), reps)
    out["walk_eval"] = timeit(lambda: eng._episode_eval(jobs, dst, sp), reps)
    vedge_dst = eng.node_vedge.gather(1, dst)
    out["critic"] = timeit(lambda: ext.critic(
        rl.contiguous(), nhop.contiguous(), vedge_dst.contiguous(),
        jobs.mask, jobs.rates.contiguous(), jobs.ul.contiguous(),
        jobs.dl.contiguous(), eng.k_conf_indptr, eng.k_conf_base,
        eng.k_conf_cols, eng.link_rates.contiguous(),
        eng.bw_comp.contiguous(), eng.k_E_arr, eng.T_arr.contiguous(),
        eng.Ee, eng.fp_iters, 0.0), reps)

    def full_step():
        jb = eng.sample_jobs(0.15, gen)
        r = eng.gnn_episode(jb, train=True)
        return r

    out["full_train_step"] = timeit(full_step, max(reps // 5, 5))
    for k, v in out.items():
        print(json.dumps({"config": name, "stage": k, "us": round(v, 1),
                          "B": eng.B, "N": eng.N, "E": eng.E,
                          "Ee": eng.Ee}), flush=True)
    return out


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="both",
                    choices=["flagship", "er1000", "both"])
    ap.add_argument("--reps", type=int, default=50)
    args = ap.parse_args()
    if args.config in ("flagship", "both"):
        bench_config("flagship_b1024_n110", 110, 1024, "ba", 16, args.reps)
    if args.config in ("er1000", "both"):
        bench_config("er1000_b64", 1000, 64, "er", 1, max(args.reps // 5, 5))
