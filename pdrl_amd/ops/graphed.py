"""Whole-step hipGraph capture for the eager-autograd algorithms
(V-MPO, SAC, SAC-Continuous, PPO-Continuous).

IMPALA/PPO run the hand-fused kernel DAG (fused_step.py); the remaining
algorithms run eager PyTorch math through the fused LSTM kernels — correct,
but a few hundred launches per step, so launch-overhead bound. This wrapper
captures ONE full updater.step (forward(s), losses, backward(s), fused
optimizers, soft updates) into a torch.cuda.CUDAGraph (hipGraph on ROCm)
and replays it.

Capture preconditions the updaters satisfy:
* all stats stay device tensors (no host syncs);
* flat fused optimizers (param/grad/state buffers are static);
* V-MPO's dual coefficient is sampled on device (RNG state advances
  correctly under graph replay);
* soft_update uses cached DEVICE pointer tables (no H2D inside the step).

If capture fails (op unsupported under capture), the step runs
stream-ordered with a one-time warning — same kernels, same numerics.
"""
from __future__ import annotations

import warnings

import torch


class GraphedUpdater:
    """Proxy around an updater: first call captures, later calls copy the
    batch into the captured static buffers and replay."""

    def __init__(self, updater):
        self._u = updater
        self._graph = None
        self._static: dict[str, torch.Tensor] | None = None
        self._stats: dict | None = None
        self._failed = False

    def __getattr__(self, name):
        return getattr(self._u, name)

    def _try_capture(self, batch):
        self._static = {k: v.detach().clone() for k, v in batch.items()}
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):  # allocator + state warmup
                    self._u.step(self._static)
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._stats = self._u.step(self._static)
            self._graph = g
        except Exception as exc:
            warnings.warn(
                f"hipGraph capture of {type(self._u).__name__}.step failed "
                f"({exc}); running stream-ordered"
            )
            self._failed = True
            self._static = None
            self._stats = None

    def step(self, batch) -> dict:
        if self._graph is None and not self._failed:
            self._try_capture(batch)
        if self._graph is None:
            return self._u.step(batch)
        for k, v in self._static.items():
            src = batch.get(k)
            if src is not None and src.data_ptr() != v.data_ptr():
                v.copy_(src, non_blocking=True)
        self._graph.replay()
        self._u.update_count += 1
        return dict(self._stats)


def maybe_graph(updater, device) -> object:
    """Wrap eager-path updaters in whole-step graph capture on GPU."""
    import os

    from pdrl_amd import ops

    if torch.device(device).type != "cuda" or not ops.available():
        return updater
    if getattr(updater, "fused_step", None) is not None:
        return updater  # already a hand-fused DAG
    if not int(os.environ.get("PDRL_GRAPH_EAGER", "1")):
        return updater
    return GraphedUpdater(updater)
