#!/usr/bin/env python3
"""Stage ablation of cheb_bwd (timing only — masked outputs are wrong):
times the kernel with cumulative stage masks and differences them.
bit0 act-mask+db, bit1 load_acts+wgrads, bit2 dx gemm, bit3 spmv.

Run on MI355X:  python profiles/ablate_cheb_bwd.py
"""
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
    return s.elapsed_time(e) * 1000.0 / reps


def main():
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.harness.train_batched import \
        build_training_cases
    from multihop_offload_amd.ops import dispatch

    cases = build_training_cases(110, 1024, 16, 1000, 7, workers=8)
    model = ChebConvStack(K=2, dtype=torch.float32, seed=3)
    with torch.no_grad():
        for p in model.parameters():
            p.mul_(0.01)
        model.layers[-1].bias.fill_(0.5)
    eng = EpisodeEngine(cases, model, device="cuda", dtype=torch.float32)
    ext = dispatch.require_hip()
    gen = torch.Generator(device="cuda")
    gen.manual_seed(0)
    jobs = eng.sample_jobs(0.15, gen)
    params = []
    for layer in eng.model.layers:
        params += [layer.weight, layer.bias]
    x = torch.randn(eng.B, eng.Ee, 4, device="cuda")
    Wp = x.new_zeros(5, 2, 32, 32)
    bp = x.new_zeros(5, 32)
    for l in range(5):
        w, b = params[2 * l], params[2 * l + 1]
        Wp[l, :w.shape[0], :w.shape[1], :w.shape[2]] = w
        bp[l, :b.shape[0]] = b
    lam, acts, t1s = ext.cheb_fwd(x, Wp, bp, eng.k_ext_indptr,
                                  eng.k_ext_base, eng.k_ext_cols,
                                  eng.k_ext_max_nnz)
    dlam = torch.randn_like(lam)

    def run(mask):
        return timeit(lambda: ext.cheb_bwd_ablate(
            dlam, acts, t1s, Wp, eng.k_ext_indptr, eng.k_ext_base,
            eng.k_ext_cols, eng.k_ext_max_nnz, mask))

    t = {m: run(m) for m in (0, 1, 3, 7, 15)}
    out = {
        "skeleton(us)": round(t[0], 1),
        "act_mask_db(us)": round(t[1] - t[0], 1),
        "load_acts_wgrads(us)": round(t[3] - t[1], 1),
        "dx_gemm(us)": round(t[7] - t[3], 1),
        "spmv(us)": round(t[15] - t[7], 1),
        "full(us)": round(t[15], 1),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
