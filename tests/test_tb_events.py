"""Native TensorBoard event-file writer (utils/tb_events.py): TFRecord
framing with masked CRC32C + hand-encoded Event protos. Validated by
round-tripping through an independent parser and against known CRC32C
vectors (reference logging surface: agents/learner.py:77-79, 95-158)."""
import glob
import struct

from pdrl_amd.utils.logger import SummaryWriter, read_scalars
from pdrl_amd.utils.tb_events import (
    EventFileWriter, _crc32c, frame_record, parse_scalar, read_records,
    scalar_event,
)


def test_crc32c_known_vectors():
    # RFC 3720 / kernel test vectors for CRC32C (Castagnoli)
    assert _crc32c(b"") == 0x00000000
    assert _crc32c(b"123456789") == 0xE3069283
    assert _crc32c(b"\x00" * 32) == 0x8A9136AA


def test_event_file_roundtrip(tmp_path):
    w = EventFileWriter(str(tmp_path))
    rows = [("loss", 0.5, 1), ("loss", 0.25, 2), ("reward", 123.0, 2)]
    for tag, v, s in rows:
        w.add_scalar(tag, v, s)
    w.close()

    files = glob.glob(str(tmp_path / "events.out.tfevents.*"))
    assert len(files) == 1
    records = read_records(files[0])  # checksums verified inside
    assert len(records) == 1 + len(rows)
    assert parse_scalar(records[0]) is None  # file_version record
    got = [parse_scalar(r) for r in records[1:]]
    for (tag, v, s), (gtag, gv, gs) in zip(rows, got):
        assert gtag == tag and gs == s
        assert abs(gv - v) < 1e-6


def test_frame_record_layout():
    payload = scalar_event("t", 1.0, 7, wall=123.0)
    rec = frame_record(payload)
    (n,) = struct.unpack("<Q", rec[:8])
    assert n == len(payload)
    assert rec[12:12 + n] == payload
    assert len(rec) == 16 + n


def test_summary_writer_emits_both(tmp_path):
    w = SummaryWriter(str(tmp_path))
    w.add_scalar("a", 1.5, 0)
    w.add_scalar("a", 2.5, 1)
    w.flush()
    w.close()
    scal = read_scalars(str(tmp_path))
    assert scal["a"] == [(0, 1.5), (1, 2.5)]
    files = glob.glob(str(tmp_path / "events.out.tfevents.*"))
    assert len(files) == 1
    got = [parse_scalar(r) for r in read_records(files[0])[1:]]
    assert got == [("a", 1.5, 0), ("a", 2.5, 1)]


def test_varint_large_step_roundtrip(tmp_path):
    """Steps beyond 2^32 must survive the varint encoding."""
    from pdrl_amd.utils.tb_events import (EventFileWriter, parse_scalar,
                                          read_records)
    import glob

    w = EventFileWriter(str(tmp_path))
    big = (1 << 40) + 12345
    w.add_scalar("x", -1.0, big)
    w.close()
    f = glob.glob(str(tmp_path / "events.out.tfevents.*"))[0]
    tag, v, s = parse_scalar(read_records(f)[1])
    assert (tag, s) == ("x", big)
    assert abs(v + 1.0) < 1e-6
