"""Worker: CPU rollout collector.

Capability parity with the reference's agents/worker.py: owns an env + the
actor module on CPU in eval mode (worker.py:32); steps the env with
``model.act`` building a per-step record of 10 fields + episode uuid
(110-123); publishes each step over the data plane (58-60); hot-reloads
actor weights from the learner's weight plane (62-72); publishes episode
reward stats at episode end (74-78); stamps a heartbeat each step (133-134).

Differences by design:
* one loop interleaving collection and weight polling (the threaded transport
  makes the reference's two asyncio tasks unnecessary);
* the reference's hard-coded ``asyncio.sleep(0.05)`` per-step throttle
  (worker.py:131 — capping every worker at 20 steps/s) is a config knob
  ``worker_step_sleep`` defaulting to 0.
"""
from __future__ import annotations

import time
import uuid

import torch

from pdrl_amd.agents.env_maker import EnvBase
from pdrl_amd.transport import pub_connect, sub_connect
from pdrl_amd.utils import Protocol, encode, decode


class Worker:
    def __init__(
        self,
        model,
        worker_idx: int,
        manager_ip: str,
        manager_port: int,
        learner_ip: str,
        learner_port: int,
        params,
        heartbeat=None,
        stop_event=None,
        seed: int | None = None,
    ):
        self.params = params
        self.worker_idx = worker_idx
        self.model = model.cpu().eval()
        # M envs per worker process (default 1 = reference behavior): one
        # batched model.act per tick steps all of them, amortizing the
        # Python/inference overhead that bounds per-worker throughput.
        self.num_envs = max(1, int(getattr(params, "num_envs_per_worker", 1) or 1))
        self.envs = [
            EnvBase(params.env, seed=None if seed is None else seed * 1000 + i)
            for i in range(self.num_envs)
        ]
        self.env = self.envs[0]
        self.heartbeat = heartbeat
        self.stop_event = stop_event
        # data plane: PUB → manager; weight plane: SUB ← learner (port+1)
        self.pub = pub_connect(manager_ip, manager_port)
        self.sub = sub_connect(learner_ip, learner_port + 1)
        self.step_sleep = float(getattr(params, "worker_step_sleep", 0.0))
        # optional ingest throttle (env-steps/s per worker, 0 = unthrottled):
        # the C++ actor plane is fast enough that FULL-rate fleets can push
        # on-policy algorithms into policy-lag instability (see
        # profiles/training_runs/README.md) — this caps the rate directly
        # instead of forcing fleet resizing
        self.max_rate = float(getattr(params, "worker_max_steps_per_sec",
                                      0.0))
        # steps per wire message: one decode per chunk instead of per step
        # (a per-step wire protocol capped the single storage process at
        # ~10K steps/s; seq_len-sized chunks amortize pickle+zlib 5×)
        self.batch_steps = int(getattr(params, "worker_batch_steps", 0)) or params.seq_len
        self._step_buf: list[dict] = []
        # Warmup exploration for continuous control (off by default = reference
        # behavior). Per-step policy noise is temporally UNCORRELATED, which
        # cannot build momentum on tasks like MountainCarContinuous; for the
        # first ``explore_warmup_steps`` env steps actions come from an
        # Ornstein-Uhlenbeck process instead (correlated bang-bang), with the
        # record's log_prob still evaluated under the CURRENT policy.
        self.explore_warmup_steps = int(getattr(params, "explore_warmup_steps", 0))
        # dedicated explorers: workers with idx < explore_ou_workers use OU
        # actions FOREVER, keeping fresh success data in the replay even
        # after the trained policy takes over the rest of the fleet
        if worker_idx < int(getattr(params, "explore_ou_workers", 0) or 0):
            self.explore_warmup_steps = 1 << 62
        if self.explore_warmup_steps:
            from pdrl_amd.agents.learner_module import is_on_policy

            # off-policy only: PPO-C/V-MPO importance ratios assume actions
            # came from the recorded policy, which OU actions do not
            if is_on_policy(getattr(params, "algo", "")):
                self.explore_warmup_steps = 0
        self._ou_theta = float(getattr(params, "explore_ou_theta", 0.15))
        self._ou_sigma = float(getattr(params, "explore_ou_sigma", 0.6))
        self._continuous = bool(getattr(params, "continuous", False))
        self._total_steps = 0
        self._ou_state: torch.Tensor | None = None
        self._ou_gen = torch.Generator().manual_seed((seed or 0) * 9973 + 17)
        # C++ batched act (one native call per tick) when the actor is a
        # standard discrete single-body core and the extension is built;
        # otherwise the eager model.act path runs (continuous policies,
        # missing extension)
        self._act = self._make_fast_act(seed) or self.model.act

    def _make_fast_act(self, seed):
        """One-call C++ act (body+LSTM+logits+sample): removes the ~15
        small-op eager dispatches per tick that bound worker throughput
        (ops/csrc/cpu_actor.cpp). Reads the live parameter tensors, so
        weight hot-reloads (in-place load_state_dict) apply immediately."""
        try:
            from pdrl_amd.ops import _cpu_actor
        except ImportError:
            return None
        actor = getattr(self.model, "actor", self.model)
        core = getattr(actor, "core", None)
        if core is None or getattr(core, "input2_dim", None) is not None:
            return None
        heads = list(core.head_names or [])
        rng = torch.tensor([((seed or 0) * 0x9E3779B97F4A7C15 + 0x2545F491) &
                            ((1 << 63) - 1) | 1], dtype=torch.int64)
        # captured ONCE: in-place load_state_dict keeps the same storages,
        # so hot-reloads still apply; the C++ kernel only reads data_ptr
        # (per-call .detach() cost ~12% of the tick in profiling)
        wargs = (core.body_w, core.body_b, core.w_ih, core.w_hh,
                 core.b_g, core.heads_w, core.heads_b)

        if not self._continuous and heads and heads[0] == "logits":
            A = core.head_dims["logits"]

            def fast(obs, hxs, outs=None):
                hx, cx = hxs
                a, lg, lp, h, c = _cpu_actor.act_batch_discrete(
                    obs.contiguous(), hx.contiguous(), cx.contiguous(),
                    *wargs, A, rng, outs=outs)
                return a, lg, lp, (h, c)

            fast.out_shapes = lambda M, H: [
                (torch.int64, (M, 1)), (torch.float32, (M, A)),
                (torch.float32, (M, 1)), (torch.float32, (M, H)),
                (torch.float32, (M, H))]
            return fast
        if self._continuous and heads[:2] in (["mu", "std"], ["mu", "log_std"]):
            A = core.head_dims["mu"]
            mode = 0 if heads[1] == "std" else 1  # PPO-C | SAC-C sampling

            def fastc(obs, hxs, outs=None):
                hx, cx = hxs
                a, lg, lp, h, c = _cpu_actor.act_batch_gaussian(
                    obs.contiguous(), hx.contiguous(), cx.contiguous(),
                    *wargs, A, mode, rng, outs=outs)
                return a, lg, lp, (h, c)

            fastc.out_shapes = lambda M, H: [
                (torch.float32, (M, A)), (torch.float32, (M, 2 * A)),
                (torch.float32, (M, 1)), (torch.float32, (M, H)),
                (torch.float32, (M, H))]
            return fastc
        return None

    # ------------------------------------------------------------------ #
    def _ou_explore(self, action, logits):
        """OU-noise action + its tanh-Gaussian log-prob under the policy."""
        if self._ou_state is None:
            self._ou_state = torch.zeros_like(action, dtype=torch.float32)
        noise = torch.randn(action.shape, generator=self._ou_gen)
        self._ou_state = (
            self._ou_state - self._ou_theta * self._ou_state + self._ou_sigma * noise
        )
        a = torch.tanh(self._ou_state)
        A = action.shape[-1]
        mu, log_std = logits[..., :A], logits[..., A:]
        std = log_std.clamp(-20.0, 2.0).exp()
        z = self._ou_state
        log_prob = (
            -0.5 * ((z - mu) / std) ** 2 - log_std - 0.5 * torch.log(torch.tensor(2 * torch.pi))
            - torch.log(1.0 - a.pow(2) + 1e-7)
        ).sum(-1, keepdim=True)
        return a, log_prob

    # ------------------------------------------------------------------ #
    def pub_rollout(self, step_data: dict, flush: bool = False):
        """Buffer the step; publish a chunk every ``batch_steps`` steps and at
        episode end (the storage-side assembler consumes chunks step-wise, so
        assembly semantics are unchanged)."""
        self._step_buf.append(step_data)
        if len(self._step_buf) >= self.batch_steps or flush:
            from pdrl_amd.buffers.wire import pack_steps

            header, payload = encode(Protocol.Rollout, pack_steps(self._step_buf),
                                     compress=False)
            self.pub.send(header, payload)
            self._step_buf = []

    def pub_stat(self, epi_rew: float):
        header, payload = encode(Protocol.Stat, {"epi_rew": float(epi_rew)})
        self.pub.send(header, payload)

    def poll_model(self):
        msg = self.sub.recv(timeout=0.0)
        # drain to the newest weight broadcast
        newest = None
        while msg is not None:
            newest = msg
            msg = self.sub.recv(timeout=0.0)
        if newest is not None:
            protocol, state_dict = decode(*newest)
            if protocol is Protocol.Model:
                from pdrl_amd.buffers.wire import is_packed_weights, unpack_weights

                if is_packed_weights(state_dict):
                    state_dict = unpack_weights(state_dict)
                actor = getattr(self.model, "actor", self.model)
                actor.load_state_dict(state_dict)

    def _stopped(self) -> bool:
        return self.stop_event is not None and self.stop_event.is_set()

    # ------------------------------------------------------------------ #
    def collect(self, max_episodes: int | None = None):
        """Roll out episodes forever (or for max_episodes, for tests).

        All env counts route through the vectorized loop (packed-chunk
        records + C++ batched physics — at M=1 it is simply faster than
        the per-step dict path below, which is kept for reference/debug
        via PDRL_SCALAR_WORKER=1)."""
        import os

        if self.num_envs > 1 or not bool(
                int(os.environ.get("PDRL_SCALAR_WORKER", "0"))):
            return self._collect_vec(max_episodes)
        H = self.params.hidden_size
        episodes = 0
        while not self._stopped():
            obs = self.env.reset()
            hx = torch.zeros(1, H)
            cx = torch.zeros(1, H)
            self._ou_state = None  # OU noise restarts with each episode
            epi_rew = 0.0
            epi_id = uuid.uuid4().hex
            is_fir = 1.0
            for _ in range(self.params.time_horizon):
                if self._stopped():
                    break
                self.poll_model()
                action, logits, log_prob, (next_hx, next_cx) = self._act(obs, (hx, cx))
                if self._continuous and self._total_steps < self.explore_warmup_steps:
                    action, log_prob = self._ou_explore(action, logits)
                self._total_steps += 1
                next_obs, rew, done, _ = self.env.step(action)
                epi_rew += rew
                step_data = {
                    "obs": obs.squeeze(0).numpy(),
                    "act": action.reshape(-1).float().numpy(),
                    "rew": rew,
                    "logits": logits.squeeze(0).numpy(),
                    "log_prob": log_prob.reshape(-1).numpy(),
                    "is_fir": is_fir,
                    "done": float(done),
                    "hx": hx.squeeze(0).numpy(),
                    "cx": cx.squeeze(0).numpy(),
                    "id": epi_id,
                }
                self.pub_rollout(step_data, flush=done)
                obs, hx, cx = next_obs, next_hx, next_cx
                is_fir = 0.0
                if self.heartbeat is not None:
                    self.heartbeat.value = time.time()
                if self.step_sleep > 0:
                    time.sleep(self.step_sleep)
                if done:
                    break
            self.pub_stat(epi_rew)
            episodes += 1
            if max_episodes is not None and episodes >= max_episodes:
                break

    def _make_batch_env(self):
        """Vectorized native-env physics: ONE C++ call steps all M envs
        (ops/csrc/cpu_actor.cpp replicates envs/{cartpole,mountain_car}.py
        dynamics; parity-tested in tests/test_envs.py). Resets stay with
        the per-env objects (their own RNG); the caller re-syncs the state
        row after each reset. Returns None for mixed/foreign env types
        (gymnasium adapter etc.) — the per-env python loop handles those."""
        try:
            from pdrl_amd.ops import _cpu_actor
        except ImportError:
            return None
        import numpy as np

        raws = [getattr(e, "env", None) for e in self.envs]
        names = {type(r).__name__ for r in raws}
        if names == {"CartPoleEnv"}:
            fn, dim = _cpu_actor.cartpole_step_batch, 4
        elif names == {"MountainCarContinuousEnv"}:
            fn, dim = _cpu_actor.mcc_step_batch, 2
        else:
            return None
        M = len(raws)
        state = torch.empty(M, dim, dtype=torch.float64)
        for i, r in enumerate(raws):
            if getattr(r, "_state", None) is None:
                return None  # env not reset yet
            state[i] = torch.from_numpy(np.asarray(r._state))
        act_f32 = torch.empty(M, dtype=torch.float32)
        outs = [torch.empty(M, dim, dtype=torch.float32),
                torch.empty(M, dtype=torch.float32),
                torch.empty(M, dtype=torch.float32)]
        return {"fn": fn, "state": state,
                "steps": torch.zeros(M, dtype=torch.int64),
                "max_steps": int(raws[0].MAX_EPISODE_STEPS), "raws": raws,
                "act_f32": act_f32, "act_np": act_f32.numpy(),
                "outs": outs, "onp": [t.numpy() for t in outs]}

    def _collect_vec(self, max_episodes: int | None = None):
        """Vectorized rollout: M envs, ONE batched model.act per tick. Each
        env keeps its own episode uuid / recurrent-state row / reward
        accumulator; records are identical to the scalar path (the storage
        assembler routes per step by uuid, so mixed-uuid chunks are fine).

        Records are written STRAIGHT into the packed wire matrix
        (buffers/wire.py row layout) with ~9 vectorized slice assignments
        per tick — the per-step dict + pack_steps re-copy path cost more
        than the model inference once the C++ actor landed."""
        import numpy as np

        p = self.params
        H, M = p.hidden_size, self.num_envs
        obs = torch.cat([e.reset() for e in self.envs], dim=0)  # (M, F)
        benv = self._make_batch_env()  # after reset: states are live
        hx = torch.zeros(M, H)
        cx = torch.zeros(M, H)
        # ping-ponged persistent act output buffers + cached numpy views:
        # kills the 5 allocations and ~7 tensor.numpy() calls per tick
        # (the C++ act writes the new state while reading the old, so two
        # slots alternate; packing reads the INPUT state's cached views)
        slots = None
        if hasattr(self._act, "out_shapes"):
            def mk_slot():
                ts = [torch.empty(shape, dtype=dt)
                      for dt, shape in self._act.out_shapes(M, H)]
                return {"t": ts, "np": [t.numpy() for t in ts]}

            slots = (mk_slot(), mk_slot())
            cur = 0
            hx_np, cx_np = hx.numpy(), cx.numpy()
        epi_rew = [0.0] * M
        epi_id = [uuid.uuid4().hex for _ in range(M)]
        is_fir = [1.0] * M
        epi_steps = [0] * M
        episodes = 0

        chunk_buf = None  # built lazily once the record widths are known
        obs_np = None  # numpy alias of obs when the C++ env path is active
        t_start = time.perf_counter()
        while not self._stopped():
            self.poll_model()
            if slots is not None:
                slot = slots[cur]
                action, logits, log_prob, (next_hx, next_cx) = self._act(
                    obs, (hx, cx), outs=slot["t"])
            else:
                action, logits, log_prob, (next_hx, next_cx) = \
                    self._act(obs, (hx, cx))
            if self._continuous and self._total_steps < self.explore_warmup_steps:
                action, log_prob = self._ou_explore(action, logits)
            self._total_steps += M
            if chunk_buf is None:
                # wire row layout (FIELD_ORDER): obs|act|rew|logits|
                # log_prob|is_fir|done|hx|cx
                widths = [obs.shape[1], action.reshape(M, -1).shape[1], 1,
                          logits.shape[1], 1, 1, 1, H, H]
                offs = np.cumsum([0] + widths)
                cap = max(self.batch_steps, 1) + M
                chunk_buf = np.empty((cap, int(offs[-1])), dtype=np.float32)
                chunk_ids: list = []
            n0 = len(chunk_ids)
            sl = slice(n0, n0 + M)
            # packing + env stepping both consume numpy views — convert the
            # act outputs ONCE per tick (torch reshape/float per tick cost
            # ~20% of the loop in profiling)
            if slots is not None and action is slot["t"][0]:
                act_np, logits_np, logp_np = (slot["np"][0], slot["np"][1],
                                              slot["np"][2])
            else:  # OU override or eager act: fresh tensors
                act_np = action.detach().numpy()
                logits_np = logits.detach().numpy()
                logp_np = log_prob.detach().numpy()
            if benv is not None:
                acts = None  # C++ batch stepper consumes act_np directly
            elif self._continuous:
                acts = [act_np[i].astype(np.float32, copy=False)
                        for i in range(M)]
            else:
                acts = [int(a) for a in act_np.reshape(-1)]
            chunk_buf[sl, offs[0]:offs[1]] = obs_np if obs_np is not None \
                else obs.numpy()
            chunk_buf[sl, offs[1]:offs[2]] = \
                act_np.reshape(M, -1).astype(np.float32, copy=False)
            chunk_buf[sl, offs[3]:offs[4]] = logits_np
            chunk_buf[sl, offs[4]:offs[5]] = logp_np.reshape(M, -1)
            chunk_buf[sl, offs[5]] = is_fir
            chunk_buf[sl, offs[7]:offs[8]] = hx_np if slots is not None \
                else hx.numpy()
            chunk_buf[sl, offs[8]:offs[9]] = cx_np if slots is not None \
                else cx.numpy()

            any_done = False
            next_rows = []
            if benv is not None:
                np.copyto(benv["act_np"], act_np.reshape(M, -1)[:, 0],
                          casting="unsafe")
                benv["fn"](benv["state"], benv["act_f32"], benv["steps"],
                           benv["max_steps"], outs=benv["outs"])
                b_obs_np, b_rew_np, b_done_np = benv["onp"]
                # vectorized column writes (tensor/scalar indexing per env
                # measurably taxed the tick)
                chunk_buf[sl, offs[2]] = b_rew_np
                chunk_buf[sl, offs[6]] = b_done_np
            for i, env in enumerate(self.envs):
                if benv is not None:
                    rew = float(b_rew_np[i])
                    done = bool(b_done_np[i])
                    next_obs = None  # row read from b_obs_np below
                else:
                    next_obs, rew, done, _ = env.step(acts[i])
                    chunk_buf[n0 + i, offs[2]] = rew
                    chunk_buf[n0 + i, offs[6]] = float(done)
                epi_rew[i] += rew
                chunk_ids.append(epi_id[i])
                any_done = any_done or done
                epi_steps[i] += 1
                horizon = done or epi_steps[i] >= p.time_horizon
                if horizon:
                    self.pub_stat(epi_rew[i])
                    episodes += 1
                    next_obs = env.reset()
                    if benv is not None:
                        benv["state"][i] = torch.from_numpy(
                            np.asarray(benv["raws"][i]._state))
                        benv["steps"][i] = 0
                        b_obs_np[i] = next_obs.numpy()[0]
                    next_hx[i] = 0.0
                    next_cx[i] = 0.0
                    if self._ou_state is not None:
                        self._ou_state[i] = 0.0
                    epi_rew[i] = 0.0
                    epi_id[i] = uuid.uuid4().hex
                    is_fir[i] = 1.0
                    epi_steps[i] = 0
                else:
                    is_fir[i] = 0.0
                if benv is None:
                    next_rows.append(next_obs)
            n = len(chunk_ids)
            if n >= self.batch_steps or any_done:
                header, payload = encode(
                    Protocol.Rollout,
                    {"ids": chunk_ids, "widths": widths,
                     "pk": chunk_buf[:n].copy()},
                    compress=False,
                )
                self.pub.send(header, payload)
                chunk_ids = []
            obs = (benv["outs"][0] if benv is not None
                   else torch.cat(next_rows, dim=0))
            obs_np = b_obs_np if benv is not None else None
            hx, cx = next_hx, next_cx
            if slots is not None:
                hx_np, cx_np = slot["np"][3], slot["np"][4]
                cur = 1 - cur
            if self.heartbeat is not None:
                self.heartbeat.value = time.time()
            if self.step_sleep > 0:
                time.sleep(self.step_sleep)
            elif self.max_rate > 0:
                # token-bucket pacing toward max_rate env-steps/s
                target = self._total_steps / self.max_rate
                elapsed = time.perf_counter() - t_start
                if target > elapsed:
                    time.sleep(min(target - elapsed, 0.05))
            else:
                time.sleep(0)  # GIL handoff for in-process (thread) fleets
            if max_episodes is not None and episodes >= max_episodes:
                break

    def close(self):
        self.pub.close()
        self.sub.close()
