"""Data-parallel training over RCCL/xGMI (gloo on CPU).

The reference is single-process single-GPU; DP is the one
distributed axis of the rebuild: replicate (engine batch + model) per GPU,
all-reduce the actor gradients as ONE flat buffer before each optimizer
step.  The payload is tiny (~3.4k params), so the all-reduce is
latency-bound — a single fused ncclAllReduce per step (not per-tensor) is
the right shape for xGMI.
"""

from __future__ import annotations

import os
from typing import Iterable

import torch
import torch.distributed as dist


def init_from_env() -> tuple[int, int]:
    """Initialise torch.distributed from torchrun env vars.  Returns
    (rank, world_size); (0, 1) when not launched distributed."""
    if "RANK" not in os.environ or dist.is_initialized():
        return (dist.get_rank(), dist.get_world_size()) \
            if dist.is_initialized() else (0, 1)
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    if backend == "nccl":
        # modulo the visible device count so an oversubscribed probe
        # (2 ranks on 1 GPU — the multi-rank RCCL path exercised on a
        # single-GPU lease) maps both ranks onto device 0
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank))
                              % max(torch.cuda.device_count(), 1))
    return rank, dist.get_world_size()


def broadcast_params(model: torch.nn.Module, src: int = 0):
    """Initial weight broadcast so all ranks start identical.  Copies back
    IN PLACE (``vector_to_parameters`` would rebind ``p.data`` and break
    flat-buffer views, e.g. the fused optimizer's)."""
    if not (dist.is_available() and dist.is_initialized()):
        return
    flat = torch.nn.utils.parameters_to_vector(model.parameters())
    dist.broadcast(flat, src=src)
    off = 0
    with torch.no_grad():
        for p in model.parameters():
            n = p.numel()
            p.copy_(flat[off:off + n].view_as(p))
            off += n


class FlatAllreduce:
    """One preallocated flat buffer for the gradient all-reduce."""

    def __init__(self, params: Iterable[torch.nn.Parameter]):
        self.params = list(params)
        n = sum(p.numel() for p in self.params)
        p0 = self.params[0]
        self.buf = torch.zeros(n, dtype=p0.dtype, device=p0.device)

    def __call__(self, average: bool = True):
        if not (dist.is_available() and dist.is_initialized()):
            return
        off = 0
        for p in self.params:
            k = p.numel()
            if p.grad is not None:
                self.buf[off:off + k].copy_(p.grad.reshape(-1))
            else:
                self.buf[off:off + k].zero_()
            off += k
        dist.all_reduce(self.buf)
        if average:
            self.buf /= dist.get_world_size()
        off = 0
        for p in self.params:
            k = p.numel()
            if p.grad is None:
                p.grad = torch.zeros_like(p)
            p.grad.copy_(self.buf[off:off + k].reshape(p.shape))
            off += k
