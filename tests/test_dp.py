"""Data-parallel plumbing over gloo (world_size 2, CPU)."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from multihop_offload_amd.models.chebconv import ChebConvStack
    from multihop_offload_amd.parallel.dp import FlatAllreduce, broadcast_params

    torch.manual_seed(rank)        # different init per rank
    model = ChebConvStack(K=2, dtype=torch.float64, seed=rank)
    broadcast_params(model)

    # after broadcast all ranks agree
    v = torch.nn.utils.parameters_to_vector(model.parameters())
    vs = [torch.zeros_like(v) for _ in range(world)]
    dist.all_gather(vs, v)
    assert torch.equal(vs[0], vs[1])

    # gradient all-reduce averages across ranks
    for p in model.parameters():
        p.grad = torch.full_like(p, float(rank + 1))
    FlatAllreduce(model.parameters())(average=True)
    expect = (1 + 2) / 2.0
    ok = all(torch.allclose(p.grad, torch.full_like(p, expect))
             for p in model.parameters())
    q.put((rank, bool(ok)))
    dist.destroy_process_group()


def test_flat_allreduce_gloo_ws2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29617
    ps = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(120)
    results = {}
    while not q.empty():
        r, ok = q.get()
        results[r] = ok
    assert results == {0: True, 1: True}


def test_engine_rank_sharded_bench_step():
    """Single-process sanity of the bench step() composition."""
    import bench
    cases = bench.build_cases(20, 4, 2, 1000, 7)
    from multihop_offload_amd.engine import EpisodeEngine
    from multihop_offload_amd.models.chebconv import ChebConvStack
    model = ChebConvStack(K=2, dtype=torch.float64, seed=0)
    engine = EpisodeEngine(cases, model, device="cpu", dtype=torch.float64)
    gen = torch.Generator()
    gen.manual_seed(0)
    jobs = engine.sample_jobs(0.15, gen)
    res = engine.gnn_episode(jobs, train=True, gen=gen)
    assert torch.isfinite(res.tau).all()
    assert res.loss_fn is not None and np.isfinite(float(res.loss_fn))


def _trainer_worker(rank, world, port, tmpdir, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    try:
        from multihop_offload_amd.harness import train_batched
        history = train_batched.main([
            "--steps", "4", "--batch", "8", "--sizes", "20",
            "--distinct", "4", "--workers", "0", "--seed", "5",
            "--device", "cpu", "--eval_every", "2", "--eval_rounds", "1",
            "--guard_every", "2", "--save_every", "100",
            "--log_every", "100", "--model_root", tmpdir,
            "--training_set", "DP2"])
        evals = [h["eval_tau"] for h in history if "eval_tau" in h]
        q.put((rank, evals))
        dist.destroy_process_group()
    except Exception as e:         # surface failures through the queue
        q.put((rank, f"ERR {type(e).__name__}: {e}"))


def test_train_batched_gloo_ws2(tmp_path):
    """Full trainer main() at world=2 over gloo: eval/guard decisions are
    all-reduced (rank-coherent) and the run completes on both ranks."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_trainer_worker,
                      args=(r, 2, 29653, str(tmp_path), q)) for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(300)
    results = {}
    while not q.empty():
        r, val = q.get()
        results[r] = val
    assert isinstance(results.get(0), list), results
    assert isinstance(results.get(1), list), results
    # eval taus are all-reduced: both ranks saw identical values
    assert len(results[0]) == 2 and results[0] == results[1], results
    assert (tmp_path / "model_ChebConv_DP2_a5_c5_ACO_agent"
            / "cp-9999.ckpt.npz").exists()


def test_bench_distributed_ws2_cpu(tmp_path):
    """bench.py under torch.distributed.run, world=2, gloo on CPU: the
    driver's multi-GPU launch shape.  Rank 0 must print one valid JSON
    line with the contract fields and n_gpus=2."""
    import json
    import subprocess
    import sys
    env = dict(os.environ)
    env.pop("RANK", None); env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29671", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "16", "--nodes", "20",
         "--distinct", "4", "--device", "cpu"],
        capture_output=True, text=True, timeout=420, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2 and rec["steps"] == 2
    assert rec["value"] > 0 and rec["scaling"] == "weak"
    assert {"metric", "unit", "ms_per_step", "higher_is_better",
            "dtype", "data", "config"} <= set(rec)
