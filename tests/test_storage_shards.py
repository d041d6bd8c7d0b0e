"""Sharded learner-storage ingest: the manager routes rollout chunks across
N storage shard processes by originating worker connection (all chunks of an
episode uuid come from one worker, so per-peer routing keeps every uuid on
one shard's assembler), and all shards write the one lock-guarded
SharedRolloutRing. This is the scale-out path past the single-process
decode ceiling (~24K steps/s at 16 workers)."""
import socket
import threading
import time
import uuid

import numpy as np
import pytest

from pdrl_amd.agents import LearnerStorage, Manager, storage_shard_ports
from pdrl_amd.buffers import SharedRolloutRing, rollout_fields
from pdrl_amd.transport import Endpoint, pub_connect, sub_bind
from pdrl_amd.utils import Protocol, decode, encode

OBS, NACT, HID, SEQ = 4, 2, 8, 5


def _wait(cond, timeout=15.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if cond():
            return True
        time.sleep(0.01)
    return cond()


def _free_base(span: int) -> int:
    """A port p such that p..p+span are all bindable."""
    for _ in range(128):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        try:
            socks = []
            for off in range(span + 1):
                t = socket.socket()
                t.bind(("127.0.0.1", p + off))
                socks.append(t)
        except OSError:
            continue
        finally:
            for t in socks:
                t.close()
        return p
    raise RuntimeError("no free port span")


def _traj_steps(eid: str):
    steps = []
    for t in range(SEQ):
        steps.append({
            "obs": np.random.rand(OBS).astype(np.float32),
            "act": np.array([t % NACT], dtype=np.float32),
            "rew": np.array([1.0], dtype=np.float32),
            "logits": np.zeros(NACT, dtype=np.float32),
            "log_prob": np.array([-0.7], dtype=np.float32),
            "is_fir": np.array([1.0 if t == 0 else 0.0], dtype=np.float32),
            "done": np.array([0.0], dtype=np.float32),
            "hx": np.zeros(HID, dtype=np.float32),
            "cx": np.zeros(HID, dtype=np.float32),
            "id": eid,
        })
    return steps


def test_recv_with_peer_ids_are_stable_and_distinct():
    sub = Endpoint(bind=("127.0.0.1", 0))
    pubs = [Endpoint(connect=("127.0.0.1", sub.bound_port)) for _ in range(2)]
    assert _wait(lambda: sub.n_peers() == 2)
    for i, p in enumerate(pubs):
        for _ in range(3):
            p.send(*encode(Protocol.Stat, {"i": i}))
    by_peer = {}
    deadline = time.monotonic() + 10.0
    while sum(len(v) for v in by_peer.values()) < 6 and time.monotonic() < deadline:
        msg = sub.recv(timeout=1.0, with_peer=True)
        if msg is None:
            continue
        pid, header, payload = msg
        by_peer.setdefault(pid, []).append(decode(header, payload)[1]["i"])
    # two distinct peer ids, each carrying exactly one publisher's messages
    assert len(by_peer) == 2
    assert sorted(set(tuple(set(v)) for v in by_peer.values())) == [(0,), (1,)]
    # default recv shape is unchanged (header, payload)
    pubs[0].send(*encode(Protocol.Stat, {"i": 9}))
    msg = sub.recv(timeout=5.0)
    assert msg is not None and len(msg) == 2
    for p in pubs:
        p.close()
    sub.close()


def test_shard_port_convention():
    # shard 0 = data plane, +1 weights, +2 rendezvous, then extra shards
    assert storage_shard_ports(7000, 1) == [7000]
    assert storage_shard_ports(7000, 3) == [7000, 7003, 7004]


def test_manager_routes_each_worker_to_one_shard():
    base = _free_base(4)
    mgr_port = _free_base(0)
    shard_subs = [sub_bind("127.0.0.1", port)
                  for port in storage_shard_ports(base, 2)]
    mgr = Manager("127.0.0.1", mgr_port, "127.0.0.1", base, storage_shards=2)
    stop = threading.Event()
    mgr.stop_event = stop
    thr = threading.Thread(target=mgr.run, daemon=True)
    thr.start()

    workers = [pub_connect("127.0.0.1", mgr_port) for _ in range(4)]
    assert _wait(lambda: mgr.sub.n_peers() == 4)
    ids = {w: [str(uuid.uuid4()) for _ in range(3)] for w in range(4)}
    for w, pub in enumerate(workers):
        for eid in ids[w]:
            pub.send(*encode(Protocol.Rollout, _traj_steps(eid)))

    got = [[] for _ in shard_subs]  # per-shard uuid lists
    deadline = time.monotonic() + 15.0
    while sum(len(g) for g in got) < 12 and time.monotonic() < deadline:
        for k, sub in enumerate(shard_subs):
            for msg in sub.recv_many(64):
                proto, steps = decode(*msg)
                assert proto is Protocol.Rollout
                got[k].append(steps[0]["id"])
    assert sum(len(g) for g in got) == 12
    # both shards used, and every worker's uuids landed on exactly one shard
    assert all(g for g in got)
    for w in range(4):
        shards_seen = {k for k in range(2) if set(ids[w]) & set(got[k])}
        assert len(shards_seen) == 1
    stop.set()
    thr.join(3.0)
    for e in workers + shard_subs:
        e.close()
    mgr.close()


def test_sharded_ingest_end_to_end():
    """4 fake workers → manager(2 shards) → 2 LearnerStorage sharing ONE
    ring: every trajectory assembles (nothing split across shards) and
    stats reach shard 0's shared array."""
    import multiprocessing

    base = _free_base(4)
    mgr_port = _free_base(0)
    fields = rollout_fields(OBS, NACT, HID, False)
    ring = SharedRolloutRing(fields, SEQ, 64, on_policy=False)
    shared_stat = multiprocessing.Array("d", 3)

    class P:
        seq_len = SEQ

    stop = threading.Event()
    shards = []
    for k, port in enumerate(storage_shard_ports(base, 2)):
        s = LearnerStorage(ring, "127.0.0.1", port, P,
                           shared_stat=shared_stat if k == 0 else None,
                           stop_event=stop)
        shards.append(s)
    mgr = Manager("127.0.0.1", mgr_port, "127.0.0.1", base, storage_shards=2,
                  stop_event=stop)
    threads = [threading.Thread(target=s.run, daemon=True) for s in shards]
    threads.append(threading.Thread(target=mgr.run, daemon=True))
    for t in threads:
        t.start()

    workers = [pub_connect("127.0.0.1", mgr_port) for _ in range(4)]
    assert _wait(lambda: mgr.sub.n_peers() == 4)
    n_traj = 12
    for i in range(n_traj):
        workers[i % 4].send(
            *encode(Protocol.Rollout, _traj_steps(str(uuid.uuid4()))))
    # one worker also reports enough episode stats to trigger the 50-mean
    for _ in range(Manager.STAT_INTERVAL):
        workers[0].send(*encode(Protocol.Stat, {"epi_rew": 100.0}))

    assert _wait(lambda: ring._head.value == n_traj, timeout=20.0), \
        f"only {ring._head.value}/{n_traj} trajectories assembled"
    assert _wait(lambda: shared_stat[2] == 1.0, timeout=10.0)
    assert shared_stat[1] == pytest.approx(100.0)
    # both shards did real work (routing fanned the 4 workers out)
    assert all(s.n_stored > 0 for s in shards)

    out = ring.drain_new()
    assert out is not None and out["obs"].shape == (n_traj, SEQ, OBS)
    stop.set()
    for t in threads:
        t.join(3.0)
    for e in workers:
        e.close()
    mgr.close()
    for s in shards:
        s.close()
