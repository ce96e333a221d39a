"""Distributed communicator: RCCL over xGMI (or gloo on CPU).

The reference's complete collective inventory is:
gradient all-reduce, loss-scalar all-reduce, initial-weight broadcast, and
dataset sharding. This module provides those on torch.distributed, with the
MI355X-tuned piece being FlatGradAllreducer: every parameter's .grad is a
VIEW into one contiguous flat buffer, so the per-step DP all-reduce is a
single RCCL call on one ~38 MB fp32 (or ~19 MB bf16) message — the 9.5 M
parameter model is latency-bound on xGMI rings, so one fused bucket beats
per-tensor reduction (reference: implicit per-variable NCCL reduces under
MirroredStrategy, model_train_custom_loop.py:175).
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None) -> tuple[int, int]:
    """Initializes the process group from torchrun env; returns (rank, world)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world <= 1:
        return 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        dist.init_process_group(backend=backend, rank=rank,
                                world_size=world)
    return rank, world


def is_main() -> bool:
    return (not dist.is_initialized()) or dist.get_rank() == 0


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def barrier():
    if dist.is_initialized():
        dist.barrier()


def broadcast_parameters(model: torch.nn.Module, src: int = 0):
    """Initial-weight broadcast."""
    if not dist.is_initialized():
        return
    for p in model.state_dict().values():
        if isinstance(p, torch.Tensor):
            dist.broadcast(p, src=src)


def allreduce_scalar(value: float, device=None) -> float:
    """Sums a python scalar across ranks (loss logging)."""
    if not dist.is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t)
    return float(t.item())


class FlatGradAllreducer:
    """One-bucket gradient all-reduce.

    Allocates a single flat buffer covering every trainable parameter's
    gradient and points param.grad at slices of it; backward then
    accumulates in place and reduce() is one all_reduce(SUM) + scale.
    """

    def __init__(self, model: torch.nn.Module,
                 dtype: torch.dtype = torch.float32):
        self.params = [p for p in model.parameters() if p.requires_grad]
        total = sum(p.numel() for p in self.params)
        device = self.params[0].device if self.params else "cpu"
        self.flat = torch.zeros(total, dtype=dtype, device=device)
        self._views = []
        offset = 0
        for p in self.params:
            n = p.numel()
            view = self.flat[offset:offset + n].view_as(p)
            p.grad = view
            self._views.append(view)
            offset += n

    def zero_(self):
        self.flat.zero_()

    def _rebind(self):
        # Defensive: if anything reassigned p.grad (zero_grad(set_to_none),
        # a library hook), fold the detached gradient back into its flat
        # view so the bucket reduce never silently misses a tensor.
        for p, view in zip(self.params, self._views):
            if p.grad is None:
                p.grad = view
            elif p.grad.data_ptr() != view.data_ptr():
                view.copy_(p.grad)
                p.grad = view

    def reduce(self):
        self._rebind()
        if dist.is_initialized():
            dist.all_reduce(self.flat, op=dist.ReduceOp.SUM)
            self.flat.div_(dist.get_world_size())

    def grad_norm(self) -> float:
        return float(self.flat.norm())
