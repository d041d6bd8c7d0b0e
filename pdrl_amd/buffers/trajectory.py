"""Trajectory container: per-episode list of step dicts with a creation
timestamp (for staleness eviction) and a done flag.

Capability parity with the reference's buffers/trajectory.py ``Trajectory2``
(reference: buffers/trajectory.py:20-39); the reference's queue-backed
``Trajectory`` is dead code and intentionally not replicated.
"""
from __future__ import annotations

import time


class Trajectory:
    __slots__ = ("seq_len", "steps", "created", "done")

    def __init__(self, seq_len: int):
        self.seq_len = seq_len
        self.steps: list[dict] = []
        self.created = time.monotonic()
        self.done = False

    def append(self, step: dict):
        self.steps.append(step)
        self.done = bool(step.get("done", False))

    def __len__(self):
        return len(self.steps)

    @property
    def age(self) -> float:
        return time.monotonic() - self.created
