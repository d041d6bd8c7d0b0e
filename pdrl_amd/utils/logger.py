"""Scalar metrics logging (replacement for the reference's tensorboardX
SummaryWriter; reference: agents/learner.py:77-79, 95-158).

The container has no tensorboardX, so metrics are written as JSONL —
one ``{"tag":..., "value":..., "step":..., "wall":...}`` object per line under
``<result_dir>/scalars.jsonl`` — easy to tail, parse, and plot offline.
API mirrors the SummaryWriter subset the framework needs.
"""
from __future__ import annotations

import json
import os
import time
from pathlib import Path


class SummaryWriter:
    """Writes BOTH a JSONL stream (easy to tail/parse offline) and a native
    TensorBoard event file (utils/tb_events.py — hand-encoded Event protos,
    no tensorboardX needed), so ``tensorboard --logdir results/`` renders
    the curves directly, matching the reference's tooling surface."""

    def __init__(self, log_dir: str, tensorboard: bool = True):
        self.log_dir = Path(log_dir)
        self.log_dir.mkdir(parents=True, exist_ok=True)
        self._path = self.log_dir / "scalars.jsonl"
        self._f = open(self._path, "a", buffering=1)
        self._tb = None
        if tensorboard:
            from .tb_events import EventFileWriter

            self._tb = EventFileWriter(str(self.log_dir))

    def add_scalar(self, tag: str, value, step: int):
        rec = {"tag": tag, "value": float(value), "step": int(step), "wall": time.time()}
        self._f.write(json.dumps(rec) + "\n")
        if self._tb is not None:
            self._tb.add_scalar(tag, value, step)

    def flush(self):
        self._f.flush()
        os.fsync(self._f.fileno())
        if self._tb is not None:
            self._tb.flush()

    def close(self):
        try:
            self._f.close()
        except Exception:
            pass
        if self._tb is not None:
            self._tb.close()

    def __del__(self):
        self.close()


def read_scalars(log_dir: str):
    """Load logged scalars back as {tag: [(step, value), ...]}."""
    out: dict[str, list] = {}
    p = Path(log_dir) / "scalars.jsonl"
    if not p.exists():
        return out
    with open(p) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            rec = json.loads(line)
            out.setdefault(rec["tag"], []).append((rec["step"], rec["value"]))
    return out
