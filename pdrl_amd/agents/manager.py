"""Manager: per-worker-machine relay/aggregator between workers and the
learner-side storage.

Capability parity with the reference's agents/manager.py: SUB binds the
manager data port collecting worker rollout/stat messages (manager.py:30-36),
PUB connects to the learner port (38-40); rollouts are forwarded verbatim;
episode-reward stats are aggregated into a mean every ``stat_interval`` (50)
episodes before forwarding (51-83).
"""
from __future__ import annotations

import pickle
import time

import numpy as np

from pdrl_amd.transport import pub_connect, sub_bind
from pdrl_amd.utils import Protocol, decode, encode


def storage_shard_ports(learner_port: int, n_shards: int) -> list[int]:
    """Data-plane ports of the learner-storage shards. Shard 0 keeps the
    reference convention (learner_port); the weight plane is learner_port+1
    and the DP rendezvous learner_port+2, so extra shards bind
    learner_port+2+k (k >= 1)."""
    return [learner_port] + [learner_port + 2 + k for k in range(1, n_shards)]


class Manager:
    STAT_INTERVAL = 50

    def __init__(self, manager_ip, manager_port, learner_ip, learner_port, stop_event=None,
                 heartbeat=None, storage_shards: int = 1):
        self.sub = sub_bind(manager_ip, manager_port)
        self.pubs = [pub_connect(learner_ip, port)
                     for port in storage_shard_ports(learner_port, max(1, storage_shards))]
        self.pub = self.pubs[0]  # stat plane + back-compat alias
        self.stop_event = stop_event
        self.heartbeat = heartbeat
        self.stat_q: list[float] = []
        self.game_count = 0
        # worker-peer → shard. All steps of an episode uuid come from one
        # worker connection, so routing by peer keeps every uuid's chunks on
        # one shard's assembler without decoding payloads.
        self._route: dict[int, int] = {}

    def _stopped(self) -> bool:
        return self.stop_event is not None and self.stop_event.is_set()

    def _shard_of(self, peer_id: int) -> int:
        shard = self._route.get(peer_id)
        if shard is None:
            shard = len(self._route) % len(self.pubs)  # round-robin first-seen
            self._route[peer_id] = shard
        return shard

    def relay_batch(self, timeout: float = 0.5) -> int:
        """Drain and forward a burst of inbound messages. Rollouts are grouped
        per shard and queued with one send_many each (one lock + one reactor
        wake per shard per burst instead of per message). Returns the number
        of messages handled."""
        first = self.sub.recv(timeout=timeout, with_peer=True)
        if first is None:
            return 0
        msgs = [first]
        msgs.extend(self.sub.recv_many(1023, with_peer=True))
        out: list[list] = [[] for _ in self.pubs]
        for peer_id, header, payload in msgs:
            protocol = pickle.loads(header)
            if protocol is Protocol.Rollout:
                out[self._shard_of(peer_id)].append((header, payload))
            elif protocol is Protocol.Stat:
                _, data = decode(header, payload)
                self.game_count += 1
                self.stat_q.append(float(data["epi_rew"]))
                if len(self.stat_q) >= self.STAT_INTERVAL:
                    mean_rew = float(np.mean(self.stat_q))
                    self.pubs[0].send(*encode(
                        Protocol.Stat,
                        {"game_count": self.game_count, "mean_stat": mean_rew},
                    ))
                    self.stat_q.clear()
            else:
                raise AssertionError(f"unexpected protocol at manager: {protocol}")
        for shard, batch in enumerate(out):
            if batch:
                self.pubs[shard].send_many(batch)
        return len(msgs)

    def run(self):
        while not self._stopped():
            self.relay_batch(timeout=0.5)
            if self.heartbeat is not None:
                self.heartbeat.value = time.time()

    def close(self):
        self.sub.close()
        for p in self.pubs:
            p.close()
