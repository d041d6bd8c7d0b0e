"""Shared-memory trajectory ring: the storage→learner hand-off.

Replaces the reference's 8 flat unlocked ``mp.Array`` buffers + bare write
cursor (reference: agents/storage_module/shared_batch.py:19-107, the
by-convention race SURVEY.md §5 documents) with an explicit lock-guarded
ring. Multi-producer / multi-consumer: N storage shards push and N learner
ranks drain concurrently — every cursor mutation happens under the one
``mp.Lock``:

* one float32 ``mp.Array`` per trajectory field, shaped (capacity, seq, dim);
* monotonic ``head`` (total written) and ``consumed`` counters guarded by an
  ``mp.Lock`` — ownership is explicit, not conventional;
* on-policy mode: capacity == batch_size; the learner drains exactly
  ``batch`` slots once available; a full ring makes the producer drop (the
  data is stale by definition — fresher rollouts are behind it);
* off-policy (replay) mode: capacity == buffer_size; the producer overwrites
  oldest; the learner samples ``batch`` random filled slots (SAC replay,
  reference: shared_batch.py:71-72, learner.py:179-183).

The learner side reads into torch tensors via ``np.frombuffer`` views — and
on GPU the batch is staged through a pinned-host buffer + async H2D copy
(``BatchStager`` in pdrl_amd/agents/learner.py), replacing the reference's
shm→np→torch→.to(device) chain (reference: learner.py:197-233).
"""
from __future__ import annotations

import numpy as np
import torch
import torch.multiprocessing as mp

from pdrl_amd.utils import mul


class SharedRolloutRing:
    """Create in the parent; pass to child processes as a Process arg."""

    def __init__(self, fields: dict[str, int], seq_len: int, capacity: int, on_policy: bool):
        self.field_dims = dict(fields)
        self.seq_len = seq_len
        self.capacity = capacity
        self.on_policy = on_policy
        ctx = mp.get_context("spawn")  # match the framework's spawn start method
        self._arrays = {
            name: ctx.Array("f", mul((capacity, seq_len, dim)), lock=False)
            for name, dim in fields.items()
        }
        self._head = ctx.Value("q", 0, lock=False)  # total slots written
        self._consumed = ctx.Value("q", 0, lock=False)  # total slots consumed (on-policy)
        self._lock = ctx.Lock()
        self._views: dict[str, np.ndarray] | None = None

    # ------------------------------------------------------------------ #
    def _view(self, name: str) -> np.ndarray:
        if self._views is None:
            self._views = {}
        if name not in self._views:
            dim = self.field_dims[name]
            self._views[name] = np.frombuffer(self._arrays[name], dtype=np.float32).reshape(
                self.capacity, self.seq_len, dim
            )
        return self._views[name]

    # -- producer (learner-storage process) ----------------------------- #
    def put(self, traj: dict[str, torch.Tensor]) -> bool:
        """Write one stacked trajectory {field: (seq, dim)}. Returns False if
        dropped (on-policy ring full)."""
        with self._lock:
            head = self._head.value
            if self.on_policy and head - self._consumed.value >= self.capacity:
                return False
            slot = head % self.capacity
            for name in self.field_dims:
                arr = np.asarray(traj[name], dtype=np.float32)
                self._view(name)[slot] = arr.reshape(self.seq_len, self.field_dims[name])
            self._head.value = head + 1
            return True

    # -- consumer (learner process) -------------------------------------- #
    def available(self) -> int:
        with self._lock:
            if self.on_policy:
                return self._head.value - self._consumed.value
            return min(self._head.value, self.capacity)

    def ready(self, batch: int) -> bool:
        return self.available() >= batch

    def drain_batch(self, batch: int) -> dict[str, np.ndarray] | None:
        """On-policy: copy out exactly ``batch`` oldest unconsumed slots."""
        assert self.on_policy
        with self._lock:
            head, base = self._head.value, self._consumed.value
            if head - base < batch:
                return None
            idx = np.arange(base, base + batch) % self.capacity
            out = {name: self._view(name)[idx].copy() for name in self.field_dims}
            self._consumed.value = base + batch
            return out

    def drain_new(self, max_n: int | None = None) -> dict[str, np.ndarray] | None:
        """Off-policy ingest conveyor: copy out every slot written since the
        last drain (oldest lost on overwrite), advancing the consumed cursor.
        Used to feed the DEVICE-resident replay (buffers/device_replay.py)."""
        with self._lock:
            head, base = self._head.value, self._consumed.value
            if head - base > self.capacity:  # producer lapped us
                base = head - self.capacity
            n = head - base
            if max_n is not None:
                n = min(n, max_n)
            if n == 0:
                return None
            idx = np.arange(base, base + n) % self.capacity
            out = {name: self._view(name)[idx].copy() for name in self.field_dims}
            self._consumed.value = base + n
            return out

    def sample_batch(self, batch: int, rng: np.random.Generator) -> dict[str, np.ndarray] | None:
        """Off-policy: random sample of ``batch`` filled slots."""
        with self._lock:
            filled = min(self._head.value, self.capacity)
            if filled < batch:
                return None
            idx = rng.integers(0, filled, size=batch)
            return {name: self._view(name)[idx].copy() for name in self.field_dims}


def rollout_fields(obs_dim: int, n_actions: int, hidden: int, continuous: bool) -> dict[str, int]:
    """Canonical per-step field widths of one trajectory record."""
    act_dim = n_actions if continuous else 1
    logits_dim = 2 * n_actions if continuous else n_actions
    return {
        "obs": obs_dim,
        "act": act_dim,
        "rew": 1,
        "logits": logits_dim,
        "log_prob": 1,
        "is_fir": 1,
        "hx": hidden,
        "cx": hidden,
    }
