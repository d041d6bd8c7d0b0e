"""Data-parallel learner plane: flat gradient all-reduce over RCCL/xGMI.

The reference has a single-GPU learner and no torch.distributed at all
(SURVEY.md §2.5/§2.6); multi-GPU data parallelism is new in this framework.
Design, sized for MI355X topology:

* The whole model is ~170K fp32 params (~0.7 MB): ONE flat bucket.
  xGMI is 7 point-to-point links × ≈153 GB/s per GPU — a ring all-reduce of
  0.7 MB is latency-bound, so a single flat fused all-reduce per step (not
  per-tensor) is the right call; bucketing/overlap machinery would only add
  latency at this size.
* The flat buffer is persistent and grad tensors are copied in/out with two
  fused multi-tensor ops (`torch._foreach_*`), giving a fixed 3-op comm
  epilogue per step that is hipGraph-capturable.
* Averaging uses ReduceOp.AVG when available (RCCL supports it), else
  SUM + scale.
"""
from __future__ import annotations

import os
from contextlib import contextmanager
from datetime import timedelta

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None, timeout_s: float = 300.0) -> tuple[int, int]:
    """Initialize torch.distributed from torchrun-style env vars.
    Returns (rank, world_size); no-op (0, 1) when WORLD_SIZE is unset/1.

    backend "nccl" IS RCCL on ROCm builds; CPU tests use "gloo".
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        rank = int(os.environ["RANK"])
        if backend == "nccl":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size,
            timeout=timedelta(seconds=timeout_s),
        )
    return dist.get_rank(), dist.get_world_size()


@contextmanager
def distributed_env():
    rank, world = init_distributed()
    try:
        yield rank, world
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


class GradReducer:
    """Single flat-bucket gradient averaging across learner ranks."""

    def __init__(self, device=None, process_group=None):
        self.pg = process_group
        self.device = device
        self._flat: torch.Tensor | None = None
        self.enabled = dist.is_available() and dist.is_initialized() and dist.get_world_size(process_group) > 1

    def world_size(self) -> int:
        return dist.get_world_size(self.pg) if self.enabled else 1

    def all_reduce(self, grads: list[torch.Tensor]):
        """Average ``grads`` in place across ranks (one fused collective)."""
        if not self.enabled:
            return
        grads = [g for g in grads if g is not None]
        if not grads:
            return
        if len(grads) == 1 and grads[0].is_contiguous():
            # the fused optimizers hand over ONE flat grad buffer: reduce it
            # in place, no staging copies (2 launches fewer per step)
            flat = grads[0]
            if hasattr(dist.ReduceOp, "AVG") and dist.get_backend(self.pg) == "nccl":
                dist.all_reduce(flat, op=dist.ReduceOp.AVG, group=self.pg)
            else:
                dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.pg)
                flat.div_(self.world_size())
            return
        numel = sum(g.numel() for g in grads)
        if self._flat is None or self._flat.numel() < numel or self._flat.device != grads[0].device:
            self._flat = torch.empty(numel, dtype=grads[0].dtype, device=grads[0].device)
        flat = self._flat[:numel]
        # fused scatter into the flat bucket
        views = []
        off = 0
        for g in grads:
            n = g.numel()
            views.append(flat[off : off + n].view_as(g))
            off += n
        torch._foreach_copy_(views, grads)
        if hasattr(dist.ReduceOp, "AVG") and dist.get_backend(self.pg) == "nccl":
            dist.all_reduce(flat, op=dist.ReduceOp.AVG, group=self.pg)
        else:
            dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.pg)
            flat.div_(self.world_size())
        torch._foreach_copy_(grads, views)

    def broadcast_params(self, params: list[torch.Tensor], src: int = 0):
        """Synchronize initial parameters from rank ``src``."""
        if not self.enabled:
            return
        for p in params:
            dist.broadcast(p.data, src=src, group=self.pg)
