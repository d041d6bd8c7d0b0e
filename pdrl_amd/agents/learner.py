"""Learner: GPU training process.

Capability parity with the reference's agents/learner.py LearnerBase +
variants: model on device + optimizer (67-70), weight PUB on learner_port+1
broadcasting after every update (85-93, ppo/learning.py:108), metrics writer
(77-79, 95-158), on-policy whole-buffer / off-policy random sampling from
shared memory (168-233), latching replay-ready flag (385), checkpoint save
every ``model_save_interval`` (ppo/learning.py:113-119).

MI355X-first differences:
* the algorithm lives in an Updater (learner_module/*) with a pure
  ``step(batch)`` — the learner is a thin process shell;
* multi-GPU data parallelism: each learner rank (one per GPU) drains its own
  batch from the ring and all-reduces gradients over RCCL/xGMI via
  GradReducer (reference has a single-GPU learner only);
* batch staging goes through a pinned-host buffer + async H2D copy
  (BatchStager below) instead of per-field blocking copies.
"""
from __future__ import annotations

import os
import time
from pathlib import Path

import numpy as np
import torch

from pdrl_amd.agents.learner_module import is_on_policy, switch_module
from pdrl_amd.buffers import SharedRolloutRing
from pdrl_amd.transport import pub_bind
from pdrl_amd.utils import ExecutionTimer, Protocol, SummaryWriter, encode


class BatchStager:
    """Host→device batch staging: ONE packed pinned buffer → ONE async H2D
    copy → STABLE per-field device views (the K15 staging design in
    SURVEY.md §2.4; reference path was shm → per-field np copy → to_torch →
    blocking .to(device), learner.py:197-233).

    Because the device views are the same tensors every call, the fused-step
    hipGraph captures directly on them and replays with zero extra copies.
    """

    def __init__(self, device):
        self.device = torch.device(device)
        self.use_cuda = self.device.type == "cuda"
        self.copy_stream = torch.cuda.Stream(self.device) if self.use_cuda else None
        self._layout: list[tuple[str, tuple, int, int]] | None = None  # (k, shape, off, n)
        self._pinned: torch.Tensor | None = None
        self._dev: torch.Tensor | None = None
        self._views: dict[str, torch.Tensor] = {}
        self._host_views: dict[str, np.ndarray] = {}

    def _build(self, batch_np: dict[str, np.ndarray]):
        layout = []
        off = 0
        for k in sorted(batch_np):
            v = batch_np[k]
            n = int(v.size)
            layout.append((k, tuple(v.shape), off, n))
            off += n
        self._layout = layout
        self._pinned = torch.empty(off, dtype=torch.float32, pin_memory=True)
        self._dev = torch.empty(off, dtype=torch.float32, device=self.device)
        flat_host = self._pinned.numpy()
        for k, shape, o, n in layout:
            self._views[k] = self._dev[o : o + n].view(shape)
            self._host_views[k] = flat_host[o : o + n].reshape(shape)

    def stage(self, batch_np: dict[str, np.ndarray]) -> dict[str, torch.Tensor]:
        if not self.use_cuda:
            return {k: torch.from_numpy(v) for k, v in batch_np.items()}
        if self._layout is None:
            self._build(batch_np)
            self._ev = torch.cuda.Event()
            self._ev.record()
        self._ev.synchronize()  # previous H2D must be done reading pinned mem
        for k, shape, o, n in self._layout:
            self._host_views[k][...] = batch_np[k]
        # the copy must not overtake the previous step still reading _dev
        self.copy_stream.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(self.copy_stream):
            self._dev.copy_(self._pinned, non_blocking=True)
            self._ev.record(self.copy_stream)
        torch.cuda.current_stream(self.device).wait_stream(self.copy_stream)
        return dict(self._views)


class WeightPublisher:
    """Fast actor-weight broadcast: one fused device gather + ONE async D2H
    into a reusable pinned buffer → a packed-weights payload
    (buffers/wire.py pack format). Replaces the per-tensor blocking .cpu()
    walk of state_dict (7 syncs + slow tensor pickling — the measured
    staged-mode bottleneck, profiles/algo_breakdown_r02a.md)."""

    def __init__(self, actor: torch.nn.Module, device):
        self.device = torch.device(device)
        self.use_cuda = self.device.type == "cuda"
        sd = actor.state_dict(keep_vars=True)
        self.schema = [(k, tuple(v.shape)) for k, v in sd.items()]
        self.params = [v.detach() for v in sd.values()]
        self.numel = sum(p.numel() for p in self.params)
        if not self.use_cuda:
            return
        self._dev = torch.empty(self.numel, dtype=torch.float32,
                                device=self.device)
        self._views = []
        off = 0
        for p in self.params:
            n = p.numel()
            self._views.append(self._dev[off:off + n].view_as(p))
            off += n
        self._pinned = torch.empty(self.numel, dtype=torch.float32,
                                   pin_memory=True)
        self._host = self._pinned.numpy()

    def begin(self):
        """Queue the fused gather + async D2H of the current weights and
        record an event — no host sync. Pair with finish(); finish() MUST
        be called before the next begin() (single snapshot buffer)."""
        if not self.use_cuda:
            return
        torch._foreach_copy_(self._views, self.params)
        self._pinned.copy_(self._dev, non_blocking=True)
        if not hasattr(self, "_ev"):
            self._ev = torch.cuda.Event()
        self._ev.record()

    def finish(self) -> dict:
        """Wait for the begin() D2H and return the packed payload. encode()
        pickles (copies) the buffer before the next begin() reuses it."""
        if not self.use_cuda:
            return {
                "wschema": self.schema,
                "wbuf": np.concatenate(
                    [p.detach().cpu().numpy().reshape(-1) for p in self.params]
                ).astype(np.float32, copy=False),
            }
        self._ev.synchronize()
        return {"wschema": self.schema, "wbuf": self._host}

    def payload(self) -> dict:
        self.begin()
        return self.finish()


class AsyncWeightPublisher:
    """Pipelined weight broadcast: the hot loop only issues the fused
    device gather + an async D2H into one of two pinned snapshots and
    records an event; a publisher thread waits on the event, packs,
    encodes and sends. The broadcast still happens for every update —
    it just overlaps the next training step instead of stalling it
    (double-buffered; if both snapshots are in flight the publish is
    coalesced into the next one, which is drop-oldest semantics — the
    wire already has those). GPU-only; CPU callers use WeightPublisher."""

    def __init__(self, actor: torch.nn.Module, device, send_fn):
        import queue
        import threading

        self.device = torch.device(device)
        sd = actor.state_dict(keep_vars=True)
        self.schema = [(k, tuple(v.shape)) for k, v in sd.items()]
        self.params = [v.detach() for v in sd.values()]
        self.numel = sum(p.numel() for p in self.params)
        self.send_fn = send_fn
        self._slots = []
        for _ in range(2):
            dev = torch.empty(self.numel, dtype=torch.float32,
                              device=self.device)
            views = []
            off = 0
            for p in self.params:
                n = p.numel()
                views.append(dev[off:off + n].view_as(p))
                off += n
            pinned = torch.empty(self.numel, dtype=torch.float32,
                                 pin_memory=True)
            self._slots.append({
                "dev": dev, "views": views, "pinned": pinned,
                "host": pinned.numpy(), "ev": torch.cuda.Event(),
            })
        self._free = queue.Queue()
        for i in range(2):
            self._free.put(i)
        self._ready: "queue.Queue[int | None]" = queue.Queue()
        self._thread = threading.Thread(target=self._drain, daemon=True)
        self._thread.start()

    def publish(self):
        try:
            i = self._free.get_nowait()
        except Exception:
            return  # both snapshots in flight: coalesce into the next one
        s = self._slots[i]
        torch._foreach_copy_(s["views"], self.params)
        s["pinned"].copy_(s["dev"], non_blocking=True)
        s["ev"].record()
        self._ready.put(i)

    def _drain(self):
        from pdrl_amd.utils import Protocol, encode

        while True:
            i = self._ready.get()
            if i is None:
                return
            s = self._slots[i]
            s["ev"].synchronize()
            payload = {"wschema": self.schema, "wbuf": s["host"]}
            header, body = encode(Protocol.Model, payload, compress=False)
            try:
                self.send_fn(header, body)
            finally:
                self._free.put(i)

    def close(self):
        self._ready.put(None)
        self._thread.join(timeout=5.0)


class Learner:
    def __init__(
        self,
        ring: SharedRolloutRing,
        learner_ip: str,
        learner_port: int,
        params,
        device: str | None = None,
        shared_stat=None,
        stop_event=None,
        heartbeat=None,
        rank: int = 0,
        world_size: int = 1,
        grad_reducer=None,
        resume_path: str | None = None,
    ):
        self.params = params
        self.ring = ring
        self.rank = rank
        self.world_size = world_size
        self.stop_event = stop_event
        self.heartbeat = heartbeat
        self.shared_stat = shared_stat
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)

        updater_cls, model_cls = switch_module(params.algo)
        obs_dim, n_actions = params.obs_dim, params.n_actions
        model = model_cls(obs_dim, n_actions, params.seq_len, params.hidden_size)
        self.updater = updater_cls(model, params, self.device, grad_reducer=grad_reducer)
        if resume_path:
            self.updater.load(resume_path, map_location=self.device)
        # eager-path algos: whole-step hipGraph capture (any world size —
        # the RCCL collective captures with the graph; failure on a rank
        # falls back to stream-ordered with identical collective order)
        from pdrl_amd.ops.graphed import maybe_graph

        self.updater = maybe_graph(self.updater, self.device)

        self.on_policy = is_on_policy(params.algo)
        self.stager = BatchStager(self.device)
        self._rng = np.random.default_rng(1234 + rank)
        self.is_root = rank == 0

        # Off-policy on GPU: the replay lives in HBM (BASELINE configs[5]) —
        # the host ring becomes an ingest conveyor, sampling is device-side.
        self.device_replay = None
        if not self.on_policy and self.device.type == "cuda":
            from pdrl_amd.buffers.device_replay import DeviceReplay

            self.device_replay = DeviceReplay(
                ring.field_dims, ring.seq_len, params.buffer_size, self.device,
                seed=1234 + rank,
            )

        # weight plane: PUB bound at learner_port + 1 (rank 0 only)
        self.pub = pub_bind(learner_ip, learner_port + 1) if self.is_root else None
        self.weight_pub = None
        self.weight_pub_async = None
        if self.is_root:
            mods = self.updater.trainable_modules()
            m = mods.get("model") or next(iter(mods.values()))
            actor = getattr(m, "actor", m)
            if self.device.type == "cuda" and self.pub is not None:
                self.weight_pub_async = AsyncWeightPublisher(
                    actor, self.device, self.pub.send)
            else:
                self.weight_pub = WeightPublisher(actor, self.device)
        self.writer = SummaryWriter(params.result_dir) if self.is_root else None
        self.timer = ExecutionTimer(num_transition=params.seq_len * params.batch_size * world_size)
        self._replay_ready = False  # latching (reference: learner.py:385)

    # ------------------------------------------------------------------ #
    def _stopped(self) -> bool:
        return self.stop_event is not None and self.stop_event.is_set()

    def sample_ready(self) -> bool:
        b = self.params.batch_size
        if self.on_policy:
            return self.ring.available() >= b
        if self.device_replay is not None:
            new = self.ring.drain_new()
            if new is not None:
                self.device_replay.append_batch(new)
            if not self._replay_ready:
                self._replay_ready = self.device_replay.size >= b
            return self._replay_ready
        if not self._replay_ready:
            self._replay_ready = self.ring.ready(b)
        return self._replay_ready

    def next_batch(self, timeout: float = 30.0):
        """Block until a batch is available (or timeout/stop). Stages to device."""
        deadline = time.monotonic() + timeout
        b = self.params.batch_size
        while not self._stopped() and time.monotonic() < deadline:
            if self.sample_ready():
                if self.device_replay is not None:
                    batch = self.device_replay.sample(b)
                    if batch is not None:
                        return batch
                else:
                    batch_np = (
                        self.ring.drain_batch(b)
                        if self.on_policy
                        else self.ring.sample_batch(b, self._rng)
                    )
                    if batch_np is not None:
                        return self.stager.stage(batch_np)
            time.sleep(0.001)
        return None

    def publish_model(self):
        """Broadcast actor weights. The reference publishes after EVERY
        update (ppo/learning.py:108) — kept as the default; at multi-kHz
        update rates ``model_publish_interval`` bounds the encode cost
        (SURVEY.md §7 hard part (e))."""
        if self.pub is None:
            return
        interval = int(getattr(self.params, "model_publish_interval", 1) or 1)
        if interval > 1 and self.updater.update_count % interval != 0:
            return
        if self.weight_pub_async is not None:
            self.weight_pub_async.publish()
            return
        header, payload = encode(Protocol.Model, self.weight_pub.payload(),
                                 compress=False)
        self.pub.send(header, payload)

    def log_stats(self, stats: dict):
        if self.writer is None:
            return
        n = self.updater.update_count
        if n % self.params.loss_log_interval == 0:
            for k, v in stats.items():
                self.writer.add_scalar(k, v, n)
            for name, dq in self.timer.throughput_dict.items():
                if dq:
                    self.writer.add_scalar(f"{name}-tps", float(np.mean(dq)), n)
            for name, dq in self.timer.timer_dict.items():
                if dq:
                    self.writer.add_scalar(f"{name}-sec", float(np.mean(dq)), n)
            if self.shared_stat is not None and self.shared_stat[2] >= 1.0:
                self.writer.add_scalar("game-count", self.shared_stat[0], n)
                self.writer.add_scalar("50-game-mean-stat-of-epi-rew", self.shared_stat[1], n)
                self.shared_stat[2] = 0.0

    def save_checkpoint(self):
        if not self.is_root:
            return
        n = self.updater.update_count
        if n > 0 and n % self.params.model_save_interval == 0:
            model_dir = Path(self.params.model_dir)
            model_dir.mkdir(parents=True, exist_ok=True)
            self.updater.save(model_dir / f"{self.params.algo}_{n}.pt")

    # ------------------------------------------------------------------ #
    def run(self, max_updates: int | None = None):
        """The learning chain: batch → update → weight broadcast → log/save
        (reference: learner.py:267-305 + each learning.py loop)."""
        while not self._stopped():
            with self.timer.timer("learner-batching-time"):
                batch = self.next_batch()
            if batch is None:
                if max_updates is not None:
                    break
                continue
            with self.timer.timer("learner-throughput", check_throughput=True):
                stats = self.updater.step(batch)
            self.publish_model()
            self.log_stats(stats)
            self.save_checkpoint()
            if self.heartbeat is not None:
                self.heartbeat.value = time.time()
            if max_updates is not None and self.updater.update_count >= max_updates:
                break

    def close(self):
        if self.weight_pub_async is not None:
            self.weight_pub_async.close()
        if self.pub is not None:
            self.pub.close()
        if self.writer is not None:
            self.writer.close()


def find_latest_checkpoint(model_dir: str, algo: str):
    """Newest ``{algo}_{idx}.pt`` by trailing index (reference resume
    behavior: main.py:128-146, utils.py:93-98)."""
    from pdrl_amd.utils import extract_file_num

    d = Path(model_dir)
    if not d.is_dir():
        return None
    cands = sorted(d.glob(f"{algo}_*.pt"), key=extract_file_num)
    return str(cands[-1]) if cands else None
