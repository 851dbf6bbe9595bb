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
