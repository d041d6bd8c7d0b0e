"""TCP pub/sub transport tests (the framework's ZMQ replacement)."""
import time

import pytest

from pdrl_amd.transport import Endpoint
from pdrl_amd.utils import Protocol, decode, encode


def _wait(cond, timeout=15.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if cond():
            return True
        time.sleep(0.01)
    return cond()


def test_roundtrip_bind_pub_connect_sub():
    pub = Endpoint(bind=("127.0.0.1", 0))
    sub = Endpoint(connect=("127.0.0.1", pub.bound_port))
    assert _wait(lambda: pub.n_peers() == 1)
    header, payload = encode(Protocol.Stat, {"epi_rew": 3.5})
    pub.send(header, payload)
    msg = sub.recv(timeout=5.0)
    assert msg is not None
    protocol, data = decode(*msg)
    assert protocol is Protocol.Stat and data["epi_rew"] == 3.5
    pub.close()
    sub.close()


def test_roundtrip_connect_pub_bind_sub():
    """Reference topology: workers connect-PUB to a bound manager SUB."""
    sub = Endpoint(bind=("127.0.0.1", 0))
    pub = Endpoint(connect=("127.0.0.1", sub.bound_port))
    assert _wait(lambda: sub.n_peers() == 1)
    import numpy as np

    step = {"obs": np.arange(4, dtype=np.float32), "id": "abc"}
    pub.send(*encode(Protocol.Rollout, step))
    msg = sub.recv(timeout=5.0)
    protocol, data = decode(*msg)
    assert protocol is Protocol.Rollout
    np.testing.assert_array_equal(data["obs"], step["obs"])
    pub.close()
    sub.close()


def test_fanout_to_multiple_subscribers():
    pub = Endpoint(bind=("127.0.0.1", 0))
    subs = [Endpoint(connect=("127.0.0.1", pub.bound_port)) for _ in range(3)]
    assert _wait(lambda: pub.n_peers() == 3)
    pub.send(*encode(Protocol.Model, {"w": 1}))
    for s in subs:
        msg = s.recv(timeout=5.0)
        assert msg is not None and decode(*msg)[1] == {"w": 1}
    pub.close()
    for s in subs:
        s.close()


def test_many_publishers_one_subscriber():
    sub = Endpoint(bind=("127.0.0.1", 0))
    pubs = [Endpoint(connect=("127.0.0.1", sub.bound_port)) for _ in range(4)]
    assert _wait(lambda: sub.n_peers() == 4)
    for i, p in enumerate(pubs):
        p.send(*encode(Protocol.Stat, {"i": i}))
    got = set()
    for _ in range(4):
        msg = sub.recv(timeout=5.0)
        assert msg is not None
        got.add(decode(*msg)[1]["i"])
    assert got == {0, 1, 2, 3}
    sub.close()
    for p in pubs:
        p.close()


def test_recv_timeout_returns_none():
    sub = Endpoint(bind=("127.0.0.1", 0))
    t0 = time.monotonic()
    assert sub.recv(timeout=0.1) is None
    assert time.monotonic() - t0 < 2.0
    sub.close()


def test_rx_overflow_bounded_and_newest_survives():
    """A slow subscriber keeps bounded memory: bursts beyond the RX HWM are
    shed (drop-oldest) until read-pause flow control engages; draining then
    resumes delivery and the NEWEST message always arrives."""
    pub = Endpoint(bind=("127.0.0.1", 0))
    sub = Endpoint(connect=("127.0.0.1", pub.bound_port), recv_hwm=8)
    assert _wait(lambda: pub.n_peers() == 1)
    for i in range(64):
        pub.send(*encode(Protocol.Stat, {"i": i}))
    vals = []
    deadline = time.monotonic() + 20.0
    while time.monotonic() < deadline:
        msg = sub.recv(timeout=1.0)
        if msg is None:
            if vals and vals[-1] == 63:
                break
            continue
        vals.append(decode(*msg)[1]["i"])
    assert vals == sorted(vals)  # order preserved
    assert vals and vals[-1] == 63  # newest survived
    assert len(vals) <= 64
    pub.close()
    sub.close()


def test_pub_never_blocks_without_peer():
    pub = Endpoint(connect=("127.0.0.1", 1))  # nothing listening
    t0 = time.monotonic()
    for i in range(100):
        pub.send(b"h", b"p")
    assert time.monotonic() - t0 < 1.0
    pub.close()

def test_send_many_preserves_order_and_framing():
    pub = Endpoint(bind=("127.0.0.1", 0))
    sub = Endpoint(connect=("127.0.0.1", pub.bound_port))
    assert _wait(lambda: pub.n_peers() == 1)
    batch = [encode(Protocol.Stat, {"i": i}) for i in range(32)]
    pub.send_many(batch)
    vals = []
    deadline = time.monotonic() + 10.0
    while len(vals) < 32 and time.monotonic() < deadline:
        msg = sub.recv(timeout=1.0)
        if msg is not None:
            vals.append(decode(*msg)[1]["i"])
    assert vals == list(range(32))
    pub.close()
    sub.close()


def test_connect_side_reconnects_after_listener_restart():
    """A connect-mode endpoint transparently reconnects when its peer dies
    and comes back on the same port (worker behavior across a learner
    restart — the supervisor respawn path depends on this)."""
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()

    sub1 = Endpoint(bind=("127.0.0.1", port))
    pub = Endpoint(connect=("127.0.0.1", port))
    assert _wait(lambda: sub1.n_peers() == 1)
    pub.send(*encode(Protocol.Stat, {"i": 1}))
    assert sub1.recv(timeout=5.0) is not None
    sub1.close()  # listener dies
    assert _wait(lambda: pub.n_peers() == 0)

    sub2 = Endpoint(bind=("127.0.0.1", port))  # listener returns
    assert _wait(lambda: sub2.n_peers() == 1), "no reconnect"
    pub.send(*encode(Protocol.Stat, {"i": 2}))
    msg = sub2.recv(timeout=5.0)
    assert msg is not None and decode(*msg)[1]["i"] == 2
    pub.close()
    sub2.close()
