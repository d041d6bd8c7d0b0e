"""Native TensorBoard event-file writer (no tensorboardX/tensorflow).

The reference logs scalars through tensorboardX (reference:
agents/learner.py:77-79, 95-158). Neither tensorboardX nor protobuf
codegen is needed to produce files TensorBoard loads: an event file is a
sequence of TFRecords (length + masked-CRC32C framing) whose payloads are
hand-encodable `Event` protos —

    Event { 1: wall_time (double), 2: step (int64),
            3: file_version (string, first record only),
            5: Summary { 1: Value { 1: tag (string),
                                    2: simple_value (float) } } }

This module implements exactly that subset. Files are named
``events.out.tfevents.<ts>.<host>`` under the log dir, so a stock
``tensorboard --logdir`` pointed at the results directory renders the
training curves directly.
"""
from __future__ import annotations

import socket
import struct
import time
from pathlib import Path

# ----------------------------------------------------------------- crc32c
# Castagnoli CRC (reflected poly 0x82F63B78), table-driven — the TFRecord
# framing checksum. Pure python; called twice per record on short buffers.
_TABLE = []
for _i in range(256):
    _c = _i
    for _ in range(8):
        _c = (_c >> 1) ^ (0x82F63B78 if (_c & 1) else 0)
    _TABLE.append(_c)


def _crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = (crc >> 8) ^ _TABLE[(crc ^ b) & 0xFF]
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ------------------------------------------------------------- protobuf
def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _key(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _tag_double(field: int, v: float) -> bytes:
    return _key(field, 1) + struct.pack("<d", v)


def _tag_float(field: int, v: float) -> bytes:
    return _key(field, 5) + struct.pack("<f", v)


def _tag_varint(field: int, v: int) -> bytes:
    return _key(field, 0) + _varint(v)


def _tag_bytes(field: int, payload: bytes) -> bytes:
    return _key(field, 2) + _varint(len(payload)) + payload


def scalar_event(tag: str, value: float, step: int,
                 wall: float | None = None) -> bytes:
    val = _tag_bytes(1, _tag_bytes(1, tag.encode()) + _tag_float(2, value))
    return (_tag_double(1, time.time() if wall is None else wall)
            + _tag_varint(2, step) + _tag_bytes(5, val))


def version_event(wall: float | None = None) -> bytes:
    return (_tag_double(1, time.time() if wall is None else wall)
            + _tag_bytes(3, b"brain.Event:2"))


def frame_record(payload: bytes) -> bytes:
    """TFRecord framing: len(8B LE) + crc(len) + payload + crc(payload)."""
    header = struct.pack("<Q", len(payload))
    return (header + struct.pack("<I", _masked_crc(header)) + payload
            + struct.pack("<I", _masked_crc(payload)))


def read_records(path) -> list[bytes]:
    """Parse a TFRecord stream back (checksums verified) — used by tests
    and offline tooling; raises on corruption."""
    out = []
    buf = Path(path).read_bytes()
    off = 0
    while off < len(buf):
        header = buf[off:off + 8]
        (n,) = struct.unpack("<Q", header)
        (hcrc,) = struct.unpack("<I", buf[off + 8:off + 12])
        assert hcrc == _masked_crc(header), "length checksum mismatch"
        payload = buf[off + 12:off + 12 + n]
        (pcrc,) = struct.unpack("<I", buf[off + 12 + n:off + 16 + n])
        assert pcrc == _masked_crc(payload), "payload checksum mismatch"
        out.append(payload)
        off += 16 + n
    return out


def parse_scalar(payload: bytes):
    """Decode the Event subset written above → (tag, value, step) or None
    (for the version record)."""
    off = 0
    step = 0
    tag = None
    value = None
    while off < len(payload):
        key = payload[off]
        off += 1
        field, wire = key >> 3, key & 7
        if wire == 1:
            off += 8
        elif wire == 5:
            off += 4
        elif wire == 0:
            v = 0
            shift = 0
            while True:
                b = payload[off]
                off += 1
                v |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
            if field == 2:
                step = v
        elif wire == 2:
            n = 0
            shift = 0
            while True:
                b = payload[off]
                off += 1
                n |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
            sub = payload[off:off + n]
            off += n
            if field == 5:  # summary → value → {tag, simple_value}
                inner = sub
                # Summary.value (field 1, message)
                assert inner[0] == 0x0A
                ln = inner[1]
                v = inner[2:2 + ln]
                p = 0
                while p < len(v):
                    k = v[p]
                    p += 1
                    if k == 0x0A:  # tag
                        tl = v[p]
                        p += 1
                        tag = v[p:p + tl].decode()
                        p += tl
                    elif k == 0x15:  # simple_value
                        (value,) = struct.unpack("<f", v[p:p + 4])
                        p += 4
                    else:
                        raise AssertionError(f"unexpected key {k:#x}")
    if tag is None:
        return None
    return tag, value, step


class EventFileWriter:
    """Append-only TensorBoard event file for scalar curves."""

    def __init__(self, log_dir: str):
        d = Path(log_dir)
        d.mkdir(parents=True, exist_ok=True)
        name = f"events.out.tfevents.{int(time.time())}.{socket.gethostname()}"
        self._f = open(d / name, "ab", buffering=0)
        self._f.write(frame_record(version_event()))

    def add_scalar(self, tag: str, value: float, step: int):
        self._f.write(frame_record(scalar_event(tag, float(value),
                                                int(step))))

    def flush(self):
        self._f.flush()

    def close(self):
        try:
            self._f.close()
        except Exception:
            pass
