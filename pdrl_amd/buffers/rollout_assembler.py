"""RolloutAssembler: reassembles per-step worker messages into fixed-length
(seq_len) trajectories keyed by episode uuid.

Behavioral parity with the reference's buffers/rollout_assembler.py:25-83,
including its two quirky-but-load-bearing semantics:

* **Staleness eviction** — a partial (not-done) trajectory older than
  ``stale_s`` (0.5 s) is dropped, bounding policy lag
  (reference: rollout_assembler.py:51-56).
* **Done-splice** — an episode that finishes before reaching seq_len is
  parked; the next new episode's steps are appended onto the SMALLEST parked
  finished trajectory, with ``is_fir`` (is-first-step flag) forced to 1.0 at
  the splice point so the learner resets recurrent state there
  (reference: rollout_assembler.py:61-73).

Completed trajectories are stacked into ``{key: float32 array(seq, feat)}``
and pushed to an asyncio queue (numpy — the ring copies into shm; learner
tensors are built at batch-staging time).
"""
from __future__ import annotations

import asyncio
import time

import numpy as np
import torch

from .trajectory import Trajectory

REQUIRED_KEYS = {
    "obs",
    "act",
    "rew",
    "logits",
    "log_prob",
    "is_fir",
    "done",
    "hx",
    "cx",
    "id",
}


def stack_trajectory(steps: list[dict]) -> dict[str, np.ndarray]:
    """[{field: array(feat)} x seq] → {field: float32 array(seq, feat)}.

    numpy end-to-end: the only production consumer is the shared ring
    (which copies numpy into shared memory), so no torch tensors are
    created here — per-(step, field) torch.as_tensor/reshape and even the
    per-field from_numpy wrappers were the measured ingest-shard ceiling
    once the actor plane went native. Packed chunks stack their FULL rows
    once; fields are zero-copy column views of the (seq, W) matrix."""
    s0 = steps[0]
    if "_row" in s0 and all("_row" in s for s in steps):
        r0 = s0["_row"]
        mat = np.empty((len(steps), r0.shape[0]), dtype=np.float32)
        for i, s in enumerate(steps):  # cheaper than np.stack's preamble
            mat[i] = s["_row"]
        return {k: mat[:, lo:hi] for k, (lo, hi) in s0["_offs"].items()}
    out = {}
    for key in s0:
        if key in ("id", "_row", "_offs"):
            continue
        v0 = s0[key]
        if isinstance(v0, np.ndarray) and v0.dtype == np.float32:
            out[key] = np.stack([s[key] for s in steps])
        elif isinstance(v0, float):
            out[key] = np.array([[s[key]] for s in steps], dtype=np.float32)
        else:
            out[key] = np.stack(
                [np.asarray(s[key], dtype=np.float32).reshape(-1)
                 for s in steps])
    return out


class RolloutAssembler:
    def __init__(self, seq_len: int, out_queue: asyncio.Queue | None = None, stale_s: float = 0.5):
        self.seq_len = seq_len
        self.stale_s = stale_s
        self.active: dict[str, Trajectory] = {}  # partial, still-running episodes
        self.parked_done: dict[str, Trajectory] = {}  # finished but short episodes
        self.out_queue = out_queue if out_queue is not None else asyncio.Queue(1024)
        self._next_evict = 0.0  # staleness sweep is amortized (time-based)

    async def push(self, step: dict):
        await self.push_many((step,))

    async def push_many(self, steps):
        """Batch push: one coroutine per CHUNK instead of per step (the
        per-step coroutine dispatch measurably capped the storage shard's
        ingest rate once the actor plane went native). Field presence is
        asserted on the first step only — packed chunks (wire.unpack_steps)
        build every row with the same keys."""
        if not steps:
            return
        first = steps[0] if isinstance(steps, (list, tuple)) else next(iter(steps))
        if "_row" not in first:  # packed rows carry the schema by construction
            missing = REQUIRED_KEYS - set(first)
            assert not missing, f"rollout step missing fields: {missing}"

        # staleness eviction of partial trajectories (swept at most every
        # stale_s/4 — a per-push scan of the active dict measurably taxed
        # the ingest loop at tens of kHz push rates)
        now = time.monotonic()
        if now >= self._next_evict:
            self._next_evict = now + self.stale_s / 4
            for k in [k for k, tr in self.active.items() if tr.age > self.stale_s]:
                del self.active[k]

        active = self.active
        parked = self.parked_done
        seq_len = self.seq_len
        for step in steps:
            eid = step["id"]
            traj = active.get(eid)
            if traj is None:
                if parked:
                    # splice a fresh episode onto the smallest parked
                    # finished one (reference semantics)
                    smallest = min(parked, key=lambda k: len(parked[k]))
                    traj = parked.pop(smallest)
                    step = dict(step)
                    step["is_fir"] = 1.0
                    if "_row" in step:  # the stacked row must agree
                        row = step["_row"].copy()
                        lo, _hi = step["_offs"]["is_fir"]
                        row[lo] = 1.0
                        step["_row"] = row
                else:
                    traj = Trajectory(seq_len)
                active[eid] = traj

            traj.append(step)

            if len(traj) == seq_len:
                del active[eid]
                await self.out_queue.put(stack_trajectory(traj.steps))
            elif step.get("done", False):
                active.pop(eid, None)
                parked[eid] = traj

    async def pop(self) -> dict[str, np.ndarray]:
        return await self.out_queue.get()

    def qsize(self) -> int:
        return self.out_queue.qsize()
