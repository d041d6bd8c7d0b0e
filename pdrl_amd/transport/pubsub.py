"""TCP publish/subscribe transport for the actor plane.

The reference framework moves rollouts/stats/weights over ZeroMQ PUB/SUB
sockets (reference: agents/worker.py:45-60, manager.py:30-40,
learner_storage.py:60-66, learner.py:85-93). This container has no pyzmq, so
the framework ships its own socket-level equivalent with the same semantics
the system relies on:

* PUB never blocks the producer: each peer has a bounded send queue; when a
  slow consumer falls behind, the OLDEST messages are dropped (ZMQ HWM-style).
* SUB delivers messages from all connected peers into one bounded RX queue,
  again dropping oldest on overflow.
* Either side may bind (accept many peers) or connect (one peer, with
  automatic reconnect) — matching the reference topology where workers
  connect-PUB to a bound manager SUB, and workers connect-SUB to the
  learner's bound weight PUB.
* Flow control: when an endpoint's RX queue fills, it PAUSES reading its
  sockets; TCP backpressure then stalls upstream senders, whose bounded send
  queues drop THEIR oldest — so overload sheds load at the producer (the
  cheapest place, and what flows through is the freshest data). Without
  this, an overloaded consumer burns its CPU parsing messages it will drop
  (measured: a 64-worker firehose halved the storage process's useful
  ingest rate).
* A message is a (header, payload) byte pair — the ZMQ multipart shape the
  Protocol enum + encode/decode produce.

TRUST BOUNDARY: like the reference's pickle-over-ZMQ design (reference:
utils/utils.py:244-249), message payloads are pickled — ``decode`` runs
``pickle.loads`` on bytes from any peer that can reach a bound port, which
is arbitrary code execution for anything with network access to the
cluster. Deploy ONLY on a private/trusted network segment (the reference's
machines.json topology assumes the same); bind to loopback or an internal
interface, never a public one. Rollout payloads already travel as packed
float matrices (buffers/wire.py) — the pickle surface is the small header
and the weight dict.

Design: ONE epoll/selector reactor thread per Endpoint handles accept,
reads, writes and reconnects for ALL peers (non-blocking sockets, framed
parsing). A thread-per-peer version measured catastrophic GIL convoy on a
256-core learner machine: a 32-worker manager fell from 19K msg/s (4 peers)
to 856 msg/s (32 peers); the reactor keeps per-endpoint thread count at 1.
"""
from __future__ import annotations

import errno
import os
import selectors
import socket
import struct
import threading
import time
from collections import deque

_HDR = struct.Struct("<II")  # header_len, payload_len
_MAX_FRAME = 1 << 28
_CHUNK = 1 << 18


class _Peer:
    __slots__ = ("sock", "rx", "tx", "inflight", "inflight_off", "alive",
                 "registered", "pid")

    def __init__(self, sock: socket.socket, pid: int):
        self.sock = sock
        self.pid = pid  # stable endpoint-local id (fds are reused; this isn't)
        self.rx = bytearray()
        self.tx: deque = deque()  # droppable framed bytes objects
        self.inflight: bytes | None = None  # partially-sent frame — NEVER
        self.inflight_off = 0               # dropped (would desync framing)
        self.alive = True
        self.registered = False


class Endpoint:
    """A bind-or-connect message endpoint usable as PUB, SUB, or both.

    ``send`` fans the message out to every live peer; received messages are
    queued (bounded, drop-oldest) for ``recv``.
    """

    def __init__(
        self,
        bind: tuple[str, int] | None = None,
        connect: tuple[str, int] | None = None,
        recv_hwm: int = 8192,
        send_hwm: int = 4096,
    ):
        assert (bind is None) != (connect is None), "exactly one of bind/connect"
        self._sel = selectors.DefaultSelector()
        self._peers: dict[int, _Peer] = {}
        self._peers_lock = threading.Lock()
        self._rx: deque = deque(maxlen=recv_hwm)
        self._rx_cv = threading.Condition()
        self._send_hwm = send_hwm
        self._closed = False
        # read-pause flow control thresholds
        self._rx_high = max(2, int(recv_hwm * 0.9))
        self._rx_low = recv_hwm // 2
        self._paused = False
        self._resume_req = False
        self._listener: socket.socket | None = None
        self._peer_seq = 0
        # diagnostic counters (reactor-thread writes; reads are racy-but-fine)
        self._tx_dropped = 0  # frames shed from peer send queues (HWM)
        self._rx_dropped = 0  # frames shed from the RX deque (HWM)
        self._rx_total = 0
        self._connect_addr = connect
        self._connecting: socket.socket | None = None
        self._next_connect = 0.0

        # wake pipe: lets send()/close() interrupt the selector wait
        self._wake_r, self._wake_w = os.pipe()
        os.set_blocking(self._wake_r, False)
        self._pending_tx: deque = deque()  # frames queued by send()
        self._pending_lock = threading.Lock()

        if bind is not None:
            self._listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            self._listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            self._listener.bind(bind)
            self.bound_port = self._listener.getsockname()[1]
            self._listener.listen(1024)
            self._listener.setblocking(False)
            self._sel.register(self._listener, selectors.EVENT_READ, "accept")
        self._sel.register(self._wake_r, selectors.EVENT_READ, "wake")
        self._io = threading.Thread(target=self._io_loop, daemon=True)
        self._io.start()

    # -- reactor ------------------------------------------------------------ #
    def _wake(self):
        try:
            os.write(self._wake_w, b"x")
        except OSError:
            pass

    def _register_peer(self, sock: socket.socket):
        sock.setblocking(False)
        try:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        except OSError:
            pass
        self._peer_seq += 1
        peer = _Peer(sock, self._peer_seq)
        with self._peers_lock:
            self._peers[sock.fileno()] = peer
        self._update_interest(peer)

    def _drop_peer(self, peer: _Peer):
        peer.alive = False
        if peer.registered:
            try:
                self._sel.unregister(peer.sock)
            except (KeyError, ValueError):
                pass
            peer.registered = False
        with self._peers_lock:
            self._peers.pop(peer.sock.fileno(), -1)
        try:
            peer.sock.close()
        except OSError:
            pass

    def _update_interest(self, peer: _Peer):
        # selectors require events != 0: a fully-muted peer (paused, nothing
        # to send) is UNREGISTERED — registering it write-only would make
        # select() always-ready and spin the reactor at 100% CPU
        want = 0 if self._paused else selectors.EVENT_READ
        if peer.tx or peer.inflight is not None:
            want |= selectors.EVENT_WRITE
        try:
            if want == 0:
                if peer.registered:
                    self._sel.unregister(peer.sock)
                    peer.registered = False
            elif peer.registered:
                self._sel.modify(peer.sock, want, peer)
            else:
                self._sel.register(peer.sock, want, peer)
                peer.registered = True
        except (KeyError, ValueError, OSError):
            pass

    def _set_paused(self, paused: bool):
        if paused == self._paused:
            return
        self._paused = paused
        with self._peers_lock:
            peers = list(self._peers.values())
        for peer in peers:
            if peer.alive:
                self._update_interest(peer)

    def _io_loop(self):
        while not self._closed:
            # initiate reconnect if in connect mode with no live peer
            if self._connect_addr is not None and not self._peers \
                    and self._connecting is None:
                now = time.monotonic()
                if now >= self._next_connect:
                    self._start_connect()
            timeout = 0.2
            for key, events in self._sel.select(timeout):
                data = key.data
                if data == "wake":
                    try:
                        while os.read(self._wake_r, 4096):
                            pass
                    except (BlockingIOError, OSError):
                        pass
                elif data == "accept":
                    self._accept_ready()
                elif data == "connecting":
                    self._finish_connect(key.fileobj)
                else:
                    peer = data
                    if events & selectors.EVENT_READ:
                        self._read_ready(peer)
                    if peer.alive and events & selectors.EVENT_WRITE:
                        self._write_ready(peer)
            # distribute frames queued by send() to peer tx queues
            self._flush_pending()
            if self._resume_req:
                self._resume_req = False
                self._set_paused(False)
        # teardown
        for key in list(self._sel.get_map().values()):
            if isinstance(key.data, _Peer):
                self._drop_peer(key.data)
        try:
            self._sel.close()
        except OSError:
            pass

    def _start_connect(self):
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setblocking(False)
        try:
            rc = sock.connect_ex(self._connect_addr)
        except OSError:
            sock.close()
            self._next_connect = time.monotonic() + 0.2
            return
        if rc in (0, errno.EINPROGRESS, errno.EWOULDBLOCK):
            self._connecting = sock
            self._sel.register(sock, selectors.EVENT_WRITE, "connecting")
        else:
            sock.close()
            self._next_connect = time.monotonic() + 0.2

    def _finish_connect(self, sock):
        try:
            self._sel.unregister(sock)
        except (KeyError, ValueError):
            pass
        self._connecting = None
        err = sock.getsockopt(socket.SOL_SOCKET, socket.SO_ERROR)
        if err == 0:
            self._register_peer(sock)
        else:
            sock.close()
            self._next_connect = time.monotonic() + 0.2

    def _accept_ready(self):
        while True:
            try:
                sock, _ = self._listener.accept()
            except (BlockingIOError, OSError):
                return
            self._register_peer(sock)

    def _read_ready(self, peer: _Peer):
        try:
            while True:
                chunk = peer.sock.recv(_CHUNK)
                if not chunk:
                    self._drop_peer(peer)
                    return
                peer.rx += chunk
                if len(chunk) < _CHUNK:
                    break
        except (BlockingIOError, InterruptedError):
            pass
        except OSError:
            self._drop_peer(peer)
            return
        # parse complete frames
        buf = peer.rx
        off = 0
        msgs = []
        while len(buf) - off >= _HDR.size:
            hlen, plen = _HDR.unpack_from(buf, off)
            if hlen > _MAX_FRAME or plen > _MAX_FRAME:
                self._drop_peer(peer)
                return
            total = _HDR.size + hlen + plen
            if len(buf) - off < total:
                break
            h0 = off + _HDR.size
            msgs.append((peer.pid, bytes(buf[h0 : h0 + hlen]),
                         bytes(buf[h0 + hlen : h0 + hlen + plen])))
            off += total
        if off:
            del buf[:off]
        if msgs:
            with self._rx_cv:
                overflow = len(self._rx) + len(msgs) - self._rx.maxlen
                if overflow > 0:
                    self._rx_dropped += min(overflow, len(self._rx) + len(msgs))
                self._rx_total += len(msgs)
                self._rx.extend(msgs)  # deque(maxlen) drops oldest
                self._rx_cv.notify()
                depth = len(self._rx)
            if depth >= self._rx_high:
                self._set_paused(True)  # io thread: safe to touch interests

    def _write_ready(self, peer: _Peer):
        try:
            while True:
                if peer.inflight is None:
                    if not peer.tx:
                        break
                    peer.inflight = peer.tx.popleft()
                    peer.inflight_off = 0
                sent = peer.sock.send(peer.inflight[peer.inflight_off:])
                if sent == 0:
                    break
                peer.inflight_off += sent
                if peer.inflight_off >= len(peer.inflight):
                    peer.inflight = None
                    peer.inflight_off = 0
        except (BlockingIOError, InterruptedError):
            pass
        except OSError:
            self._drop_peer(peer)
            return
        self._update_interest(peer)

    def _flush_pending(self):
        with self._pending_lock:
            if not self._pending_tx:
                return
            frames = list(self._pending_tx)
            self._pending_tx.clear()
        with self._peers_lock:
            peers = list(self._peers.values())
        for peer in peers:
            if not peer.alive:
                continue
            for f in frames:
                if len(peer.tx) >= self._send_hwm:
                    peer.tx.popleft()  # drop oldest (PUB HWM semantics);
                    # the in-flight frame lives outside tx, so framing is safe
                    self._tx_dropped += 1
                peer.tx.append(f)
            self._write_ready(peer)

    # -- API ---------------------------------------------------------------- #
    def send(self, header: bytes, payload: bytes):
        frame = _HDR.pack(len(header), len(payload)) + header + payload
        with self._pending_lock:
            self._pending_tx.append(frame)
        self._wake()

    def send_many(self, messages):
        """Queue many (header, payload) messages with ONE lock acquisition and
        ONE reactor wake — the relay fast path (per-send overhead capped a
        24-worker manager at ~25K msg/s)."""
        if not messages:
            return
        frames = [_HDR.pack(len(h), len(p)) + h + p for h, p in messages]
        with self._pending_lock:
            self._pending_tx.extend(frames)
        self._wake()

    def recv(self, timeout: float | None = None, with_peer: bool = False):
        """Pop one (header, payload) message, or None on timeout. With
        ``with_peer`` the tuple is (peer_id, header, payload) — peer_id is a
        stable endpoint-local id of the originating connection (used by the
        manager to shard rollout routing by worker)."""
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._rx_cv:
            while not self._rx:
                if self._closed:
                    return None
                remaining = None if deadline is None else deadline - time.monotonic()
                if remaining is not None and remaining <= 0:
                    return None
                self._rx_cv.wait(timeout=remaining if remaining is not None else 0.5)
            out = self._rx.popleft()
            depth = len(self._rx)
        if self._paused and depth <= self._rx_low:
            self._resume_req = True
            self._wake()
        return out if with_peer else out[1:]

    def recv_many(self, max_n: int = 1024, with_peer: bool = False) -> list:
        """Drain up to max_n queued messages without blocking."""
        out = []
        with self._rx_cv:
            while self._rx and len(out) < max_n:
                m = self._rx.popleft()
                out.append(m if with_peer else m[1:])
            depth = len(self._rx)
        if self._paused and depth <= self._rx_low:
            self._resume_req = True
            self._wake()
        return out

    def stats(self) -> dict:
        """Diagnostic counters: frames received, and frames shed by the
        send-side (tx_dropped, per-peer HWM) and receive-side (rx_dropped,
        RX HWM) bounded queues. Localizes WHERE an overloaded pipeline is
        shedding (see profiles/ingest_shards.md)."""
        return {
            "rx_total": self._rx_total,
            "rx_dropped": self._rx_dropped,
            "tx_dropped": self._tx_dropped,
        }

    def n_peers(self) -> int:
        with self._peers_lock:
            return sum(1 for p in self._peers.values() if p.alive)

    def wait_peer(self, timeout: float = 10.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if self.n_peers() > 0:
                return True
            time.sleep(0.02)
        return self.n_peers() > 0

    def close(self):
        self._closed = True
        self._wake()
        if self._listener is not None:
            try:
                self._listener.close()
            except OSError:
                pass
        with self._rx_cv:
            self._rx_cv.notify_all()
        self._io.join(timeout=2.0)
        try:
            os.close(self._wake_r)
            os.close(self._wake_w)
        except OSError:
            pass


def pub_bind(ip: str, port: int, **kw) -> Endpoint:
    return Endpoint(bind=(ip, port), **kw)


def pub_connect(ip: str, port: int, **kw) -> Endpoint:
    return Endpoint(connect=(ip, port), **kw)


sub_bind = pub_bind
sub_connect = pub_connect
