"""LearnerStorage: learner-side ingest process.

Capability parity with the reference's agents/learner_storage.py: SUB binds
the learner data port (60-66); rollout messages feed the RolloutAssembler,
stat messages update the shared stat array (77-90, 104-121); completed
seq_len trajectories are written into shared memory for the learner
(92-102, 123-159) — here into the explicitly-synchronized SharedRolloutRing
instead of the reference's unlocked flat arrays.
"""
from __future__ import annotations

import asyncio
import time

from pdrl_amd.buffers import RolloutAssembler, SharedRolloutRing
from pdrl_amd.buffers.wire import is_packed, unpack_steps
from pdrl_amd.transport import sub_bind
from pdrl_amd.utils import Protocol, decode


class LearnerStorage:
    def __init__(
        self,
        ring: SharedRolloutRing,
        learner_ip: str,
        learner_port: int,
        params,
        shared_stat=None,  # mp.Array('d', 3): [game_count, mean_rew, fresh-flag]
        stop_event=None,
        heartbeat=None,
    ):
        self.ring = ring
        self.params = params
        self.sub = sub_bind(learner_ip, learner_port)
        self.shared_stat = shared_stat
        self.stop_event = stop_event
        self.heartbeat = heartbeat
        self.assembler = RolloutAssembler(params.seq_len)
        self.n_ingested = 0
        self.n_stored = 0

    def _stopped(self) -> bool:
        return self.stop_event is not None and self.stop_event.is_set()

    # ------------------------------------------------------------------ #
    async def ingest_task(self):
        """Data-plane SUB → assembler / stat array. Messages are drained in
        batches; a Rollout payload is a CHUNK (list of steps) or one step."""
        while not self._stopped():
            msgs = self.sub.recv_many(256)
            if not msgs:
                await asyncio.sleep(0.001)
                continue
            for msg in msgs:
                protocol, data = decode(*msg)
                if protocol is Protocol.Rollout:
                    if is_packed(data):  # packed chunk: one matrix per chunk
                        steps = unpack_steps(data, lean=True)
                    else:  # plain step dict(s) — compatibility path
                        steps = data if isinstance(data, list) else [data]
                    await self.assembler.push_many(steps)
                    self.n_ingested += len(steps)
                elif protocol is Protocol.Stat:
                    if self.shared_stat is not None:
                        self.shared_stat[0] = float(data["game_count"])
                        self.shared_stat[1] = float(data["mean_stat"])
                        self.shared_stat[2] = 1.0  # fresh flag
                else:
                    raise AssertionError(f"unexpected protocol at storage: {protocol}")
            if self.heartbeat is not None:
                self.heartbeat.value = time.time()

    async def store_task(self):
        """Assembler → shared ring. Drains in bursts: one await per WAKE,
        then get_nowait until empty (per-trajectory awaits measurably
        capped the shard once decode/stacking got fast)."""
        q = self.assembler.out_queue
        while not self._stopped():
            try:
                traj = await asyncio.wait_for(q.get(), timeout=0.5)
            except asyncio.TimeoutError:
                continue
            while True:
                self.ring.put(traj)
                self.n_stored += 1
                try:
                    traj = q.get_nowait()
                except asyncio.QueueEmpty:
                    break

    async def chain(self):
        await asyncio.gather(self.ingest_task(), self.store_task())

    def run(self):
        asyncio.run(self.chain())

    def close(self):
        self.sub.close()
