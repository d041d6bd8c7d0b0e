"""Packed rollout wire format.

A rollout chunk as a list of per-step dicts costs the storage shard one
pickle object graph per step × 10 numpy arrays per dict — the decode is the
measured per-shard ingest ceiling (profiles/ingest_shards.md). Packing the
chunk into ONE (n_steps, row_width) float32 array + a uuid list makes the
payload a single numpy buffer: decode is one allocation and n×10 array
VIEWS. The storage side stays compatible with unpacked lists (tests and
third-party producers can still send plain step dicts).

Row layout (fixed field order; widths vary per env/model and ride along
once per chunk): obs | act | rew | logits | log_prob | is_fir | done |
hx | cx.
"""
from __future__ import annotations

import numpy as np

FIELD_ORDER = ("obs", "act", "rew", "logits", "log_prob", "is_fir", "done",
               "hx", "cx")

_SPAN_CACHE: dict[tuple, dict] = {}  # widths-tuple → field span map


def pack_steps(steps: list[dict]) -> dict:
    """[{field: array-like} × n] → {"ids", "widths", "pk"} (one float32 mat)."""
    first = steps[0]
    widths = [int(np.asarray(first[k], dtype=np.float32).reshape(-1).shape[0])
              for k in FIELD_ORDER]
    W = sum(widths)
    pk = np.empty((len(steps), W), dtype=np.float32)
    for i, s in enumerate(steps):
        off = 0
        row = pk[i]
        for k, w in zip(FIELD_ORDER, widths):
            row[off:off + w] = np.asarray(s[k], dtype=np.float32).reshape(-1)
            off += w
    return {"ids": [s["id"] for s in steps], "widths": widths, "pk": pk}


def unpack_steps(obj: dict, lean: bool = False) -> list[dict]:
    """Inverse of pack_steps. Field values are VIEWS into the chunk matrix
    (zero copies; the assembler stacks them into owned trajectory tensors).
    Each step also carries its FULL row (``_row``) and the shared field
    span map (``_offs``) so stack_trajectory can stack a whole trajectory
    with ONE np.stack and slice fields zero-copy."""
    pk = obj["pk"]
    widths = obj["widths"]
    key = tuple(widths)
    span = _SPAN_CACHE.get(key)
    if span is None:  # identical for every chunk of a run — build once
        offs_l = np.cumsum([0] + list(widths))
        span = {k: (int(offs_l[j]), int(offs_l[j + 1]))
                for j, k in enumerate(FIELD_ORDER)}
        _SPAN_CACHE[key] = span
    out = []
    if lean:
        # assembler fast path: it only routes on id/done (+ is_fir via the
        # row on splice) and stacks whole rows — skip the 9 per-step field
        # views (measured ~4 µs/step of pure dict building)
        done_lo = span["done"][0]
        for i, eid in enumerate(obj["ids"]):
            row = pk[i]
            out.append({"id": eid, "done": float(row[done_lo]),
                        "_row": row, "_offs": span})
        return out
    offs = np.cumsum([0] + list(widths))
    for i, eid in enumerate(obj["ids"]):
        row = pk[i]
        step = {
            k: row[offs[j]:offs[j + 1]]
            for j, k in enumerate(FIELD_ORDER)
        }
        # scalar fields ride as 1-wide vectors; consumers index/flatten anyway
        step["rew"] = float(step["rew"][0])
        step["done"] = float(step["done"][0])
        step["is_fir"] = float(step["is_fir"][0])
        step["id"] = eid
        step["_row"] = row
        step["_offs"] = span
        out.append(step)
    return out


def is_packed(payload) -> bool:
    return isinstance(payload, dict) and "pk" in payload and "ids" in payload


# --------------------------------------------------------------------------- #
# Packed weight broadcast (learner → workers, Protocol.Model)
# --------------------------------------------------------------------------- #
def pack_weights(named: "dict[str, np.ndarray]") -> dict:
    """{name: array} → {"wschema": [(name, shape)...], "wbuf": flat fp32}.

    The reference pickles a whole state_dict of tensors per broadcast
    (reference: agents/learner.py:85-93); packing makes the payload ONE
    numpy buffer — one memcpy to serialize, and the learner can fill it
    with a single async D2H instead of per-tensor blocking .cpu() copies.
    """
    schema = [(k, tuple(v.shape)) for k, v in named.items()]
    total = sum(int(np.prod(s)) for _, s in schema)
    buf = np.empty(total, dtype=np.float32)
    off = 0
    for k, v in named.items():
        n = int(np.asarray(v).size)
        buf[off:off + n] = np.asarray(v, dtype=np.float32).reshape(-1)
        off += n
    return {"wschema": schema, "wbuf": buf}


def unpack_weights(obj: dict):
    """Inverse of pack_weights → {name: torch.Tensor} (tensor views of the
    decoded buffer; load_state_dict copies them into the model)."""
    import torch

    buf = torch.from_numpy(np.asarray(obj["wbuf"], dtype=np.float32))
    out = {}
    off = 0
    for name, shape in obj["wschema"]:
        n = int(np.prod(shape)) if shape else 1
        out[name] = buf[off:off + n].view(shape)
        off += n
    return out


def is_packed_weights(payload) -> bool:
    return isinstance(payload, dict) and "wbuf" in payload and "wschema" in payload
