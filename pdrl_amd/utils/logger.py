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
    def __init__(self, log_dir: str):
        self.log_dir = Path(log_dir)
        self.log_dir.mkdir(parents=True, exist_ok=True)
        self._path = self.log_dir / "scalars.jsonl"
        self._f = open(self._path, "a", buffering=1)

    def add_scalar(self, tag: str, value, step: int):
        rec = {"tag": tag, "value": float(value), "step": int(step), "wall": time.time()}
        self._f.write(json.dumps(rec) + "\n")

    def flush(self):
        self._f.flush()
        os.fsync(self._f.fileno())

    def close(self):
        try:
            self._f.close()
        except Exception:
            pass

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
