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
* A message is a (header, payload) byte pair — the ZMQ multipart shape the
  Protocol enum + encode/decode produce.

Threads, not asyncio: the transport must be usable from plain worker loops
and from asyncio agents alike (async wrappers poll the queue).
"""
from __future__ import annotations

import socket
import struct
import threading
import time
from collections import deque

_HDR = struct.Struct("<II")  # header_len, payload_len
_MAX_FRAME = 1 << 28


class _Peer:
    """One connected remote: a writer thread draining a bounded deque."""

    def __init__(self, sock: socket.socket, on_message, send_hwm: int):
        self.sock = sock
        self.alive = True
        self._on_message = on_message
        self._q: deque = deque(maxlen=send_hwm)
        self._cv = threading.Condition()
        self._writer = threading.Thread(target=self._write_loop, daemon=True)
        self._reader = threading.Thread(target=self._read_loop, daemon=True)
        self._writer.start()
        self._reader.start()

    def send(self, header: bytes, payload: bytes):
        with self._cv:
            self._q.append((header, payload))  # deque(maxlen) drops oldest
            self._cv.notify()

    def _write_loop(self):
        try:
            while self.alive:
                with self._cv:
                    while self.alive and not self._q:
                        self._cv.wait(timeout=0.5)
                    if not self.alive:
                        return
                    header, payload = self._q.popleft()
                msg = _HDR.pack(len(header), len(payload)) + header + payload
                self.sock.sendall(msg)
        except OSError:
            pass
        finally:
            self.close()

    def _read_loop(self):
        try:
            while self.alive:
                raw = self._recv_exact(_HDR.size)
                if raw is None:
                    return
                hlen, plen = _HDR.unpack(raw)
                if hlen > _MAX_FRAME or plen > _MAX_FRAME:
                    return
                header = self._recv_exact(hlen)
                payload = self._recv_exact(plen)
                if header is None or payload is None:
                    return
                if self._on_message is not None:
                    self._on_message(header, payload)
        except OSError:
            pass
        finally:
            self.close()

    def _recv_exact(self, n: int):
        buf = bytearray()
        while len(buf) < n:
            try:
                chunk = self.sock.recv(n - len(buf))
            except OSError:
                return None
            if not chunk:
                return None
            buf += chunk
        return bytes(buf)

    def close(self):
        if self.alive:
            self.alive = False
            with self._cv:
                self._cv.notify_all()
            try:
                self.sock.close()
            except OSError:
                pass


class Endpoint:
    """A bind-or-connect message endpoint usable as PUB, SUB, or both.

    ``send`` fans the message out to every live peer; received messages are
    queued (bounded, drop-oldest) for ``recv``.
    """

    def __init__(
        self,
        bind: tuple[str, int] | None = None,
        connect: tuple[str, int] | None = None,
        recv_hwm: int = 4096,
        send_hwm: int = 4096,
    ):
        assert (bind is None) != (connect is None), "exactly one of bind/connect"
        self._peers: list[_Peer] = []
        self._peers_lock = threading.Lock()
        self._rx: deque = deque(maxlen=recv_hwm)
        self._rx_cv = threading.Condition()
        self._send_hwm = send_hwm
        self._closed = False
        self._listener = None
        if bind is not None:
            self._listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            self._listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            self._listener.bind(bind)
            self.bound_port = self._listener.getsockname()[1]
            self._listener.listen(128)
            threading.Thread(target=self._accept_loop, daemon=True).start()
        else:
            self._connect_addr = connect
            threading.Thread(target=self._connect_loop, daemon=True).start()

    # -- wiring ------------------------------------------------------------ #
    def _on_message(self, header: bytes, payload: bytes):
        with self._rx_cv:
            self._rx.append((header, payload))
            self._rx_cv.notify()

    def _add_peer(self, sock: socket.socket):
        sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        peer = _Peer(sock, self._on_message, self._send_hwm)
        with self._peers_lock:
            self._peers = [p for p in self._peers if p.alive] + [peer]

    def _accept_loop(self):
        while not self._closed:
            try:
                sock, _ = self._listener.accept()
            except OSError:
                return
            self._add_peer(sock)

    def _connect_loop(self):
        while not self._closed:
            with self._peers_lock:
                have_live = any(p.alive for p in self._peers)
            if not have_live:
                try:
                    sock = socket.create_connection(self._connect_addr, timeout=2.0)
                    sock.settimeout(None)
                    self._add_peer(sock)
                except OSError:
                    time.sleep(0.2)
                    continue
            time.sleep(0.2)

    # -- API ---------------------------------------------------------------- #
    def send(self, header: bytes, payload: bytes):
        with self._peers_lock:
            peers = [p for p in self._peers if p.alive]
        for p in peers:
            p.send(header, payload)

    def recv(self, timeout: float | None = None):
        """Pop one (header, payload) message, or None on timeout."""
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._rx_cv:
            while not self._rx:
                if self._closed:
                    return None
                remaining = None if deadline is None else deadline - time.monotonic()
                if remaining is not None and remaining <= 0:
                    return None
                self._rx_cv.wait(timeout=remaining if remaining is not None else 0.5)
            return self._rx.popleft()

    def n_peers(self) -> int:
        with self._peers_lock:
            return sum(1 for p in self._peers if p.alive)

    def wait_peer(self, timeout: float = 10.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if self.n_peers() > 0:
                return True
            time.sleep(0.02)
        return self.n_peers() > 0

    def close(self):
        self._closed = True
        if self._listener is not None:
            try:
                self._listener.close()
            except OSError:
                pass
        with self._peers_lock:
            for p in self._peers:
                p.close()
        with self._rx_cv:
            self._rx_cv.notify_all()


def pub_bind(ip: str, port: int, **kw) -> Endpoint:
    return Endpoint(bind=(ip, port), **kw)


def pub_connect(ip: str, port: int, **kw) -> Endpoint:
    return Endpoint(connect=(ip, port), **kw)


sub_bind = pub_bind
sub_connect = pub_connect
