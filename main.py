"""Process runner / CLI: role dispatch and process supervision.

Capability parity with the reference's main.py (Runner role dispatch via
sys.argv: 54-59, 475-508; GPU pick: 66-68; env-space probe: 82-95;
module_switcher: 99-116; manager role: 228-242; worker role with per-process
CPU model + heartbeat: 244-299; learner role allocating shared memory and
spawning storage + learner processes: 301-414; checkpoint resume: 128-146;
signal/atexit cleanup: 484-502).

Improvements over the reference by design:
* the heartbeat monitor/restarter — present but fully commented out in the
  reference (main.py:417-473) — is implemented and active: a supervised
  process whose heartbeat goes stale for >60 s is terminated and respawned;
* the learner role can run N data-parallel ranks (one per MI355X GPU) that
  all-reduce gradients over RCCL (reference: single GPU only).

CLI (identical shape to the reference):
  python main.py learner_sub_process <learner_ip> <learner_port>
  python main.py manager_sub_process <manager_ip> <learner_ip> <port> <learner_port>
  python main.py worker_sub_process <num_p> <manager_ip> <learner_ip> <port> <learner_port>
"""
from __future__ import annotations

import atexit
import os
import signal
import sys
import time
import traceback

import torch
import torch.multiprocessing as mp

from pdrl_amd import envs
from pdrl_amd.envs.base import Box
from pdrl_amd.utils import (
    Params,
    refresh_result_dirs,
    save_error_log,
    select_least_used_gpu,
)

HEARTBEAT_TIMEOUT_S = 60.0

fn_dict: dict = {}


def register(fn):
    fn_dict[fn.__name__] = fn
    return fn


# --------------------------------------------------------------------------- #
# Env probing + model construction
# --------------------------------------------------------------------------- #
def probe_env_spaces(params):
    """Fill params.obs_dim / n_actions / continuous from the env
    (reference: main.py:82-95)."""
    env = envs.make(params.env)
    obs_dim = int(env.observation_space.shape[0])
    continuous = isinstance(env.action_space, Box)
    if continuous:
        n_actions = int(env.action_space.shape[0])
    else:
        n_actions = int(env.action_space.n)
    assert not params.need_conv, "conv observation path not supported"
    if continuous:
        assert params.algo.endswith("Continuous"), (
            f"continuous env {params.env} needs a *-Continuous algo, got {params.algo}"
        )
    params.obs_dim = obs_dim
    params.n_actions = n_actions
    params.continuous = continuous
    return params


def build_model(params):
    from pdrl_amd.agents.learner_module import switch_module

    _, model_cls = switch_module(params.algo)
    return model_cls(params.obs_dim, params.n_actions, params.seq_len, params.hidden_size)


def load_actor_weights(model, params):
    """Resume actor weights from the newest checkpoint, if any
    (reference: set_model_weight, main.py:128-146)."""
    from pdrl_amd.agents import find_latest_checkpoint

    path = find_latest_checkpoint(params.model_dir, params.algo)
    if path is None:
        return None
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    state = ckpt.get("modules", {}).get("model")
    if state is not None:
        model.load_state_dict(state)
    return path


# --------------------------------------------------------------------------- #
# Spawn targets (module-level: must be picklable under spawn)
# --------------------------------------------------------------------------- #
def worker_run(model, worker_idx, manager_ip, manager_port, learner_ip, learner_port,
               params, heartbeat, stop_event):
    from pdrl_amd.agents import Worker

    torch.set_num_threads(1)  # single-step CPU inference; OMP pools only thrash
    try:
        w = Worker(model, worker_idx, manager_ip, manager_port, learner_ip, learner_port,
                   params, heartbeat=heartbeat, stop_event=stop_event, seed=worker_idx)
        w.collect()
    except Exception:
        save_error_log("worker", traceback.format_exc())
        raise


def manager_run(manager_ip, manager_port, learner_ip, learner_port, heartbeat, stop_event,
                storage_shards=1):
    from pdrl_amd.agents import Manager

    try:
        m = Manager(manager_ip, manager_port, learner_ip, learner_port,
                    stop_event=stop_event, heartbeat=heartbeat,
                    storage_shards=storage_shards)
        m.run()
    except Exception:
        save_error_log("manager", traceback.format_exc())
        raise


def storage_run(ring, learner_ip, learner_port, params, shared_stat, heartbeat, stop_event):
    from pdrl_amd.agents import LearnerStorage

    torch.set_num_threads(1)
    try:
        s = LearnerStorage(ring, learner_ip, learner_port, params,
                           shared_stat=shared_stat, stop_event=stop_event, heartbeat=heartbeat)
        s.run()
    except Exception:
        save_error_log("learner_storage", traceback.format_exc())
        raise


def learner_run(ring, learner_ip, learner_port, params, shared_stat, heartbeat, stop_event,
                rank, world_size):
    from pdrl_amd.agents import Learner, find_latest_checkpoint
    from pdrl_amd.parallel import GradReducer, init_distributed

    # Resolve the newest checkpoint HERE, at every (re)start: when the
    # Supervisor respawns a crashed learner this picks up the checkpoints
    # written during the run instead of re-initializing from scratch (a
    # frozen resume_path in the spawn args would silently reset training
    # and broadcast random weights to the fleet).
    resume_path = find_latest_checkpoint(params.model_dir, params.algo)

    if not torch.cuda.is_available():
        # CPU learner shares cores with the worker fleet: cap the OMP pool
        # (8 spinning intra-op threads measured 5.6 ms steps at 1.2 s)
        torch.set_num_threads(2)
    try:
        reducer = None
        device = None
        if world_size > 1:
            os.environ["RANK"] = str(rank)
            os.environ["LOCAL_RANK"] = str(rank)
            os.environ["WORLD_SIZE"] = str(world_size)
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", str(learner_port + 2))
            init_distributed()
            reducer = GradReducer()
            device = f"cuda:{rank}" if torch.cuda.is_available() else "cpu"
        elif torch.cuda.is_available():
            device = f"cuda:{select_least_used_gpu()}"
        lrn = Learner(ring, learner_ip, learner_port, params, device=device,
                      shared_stat=shared_stat, stop_event=stop_event, heartbeat=heartbeat,
                      rank=rank, world_size=world_size, grad_reducer=reducer,
                      resume_path=resume_path)
        if reducer is not None:
            reducer.broadcast_params(list(p for m in lrn.updater.trainable_modules().values()
                                          for p in m.parameters()))
        lrn.run()
    except Exception:
        save_error_log("learner", traceback.format_exc())
        raise


# --------------------------------------------------------------------------- #
# Supervision: live heartbeat monitor (the reference's dead code, made real)
# --------------------------------------------------------------------------- #
class Supervisor:
    def __init__(self, stop_event):
        self.stop_event = stop_event
        self.specs: dict[str, dict] = {}

    def spawn(self, name: str, target, args: tuple, heartbeat=None):
        p = mp.Process(target=target, args=args, daemon=True, name=name)
        p.start()
        self.specs[name] = {"target": target, "args": args, "heartbeat": heartbeat, "proc": p}
        return p

    def monitor(self, poll_s: float = 5.0):
        """Supervise until stop: restart processes that died or whose
        heartbeat is >60 s stale (reference spec: main.py:417-473)."""
        while not self.stop_event.is_set():
            time.sleep(poll_s)
            now = time.time()
            for name, spec in self.specs.items():
                if self.stop_event.is_set():
                    return
                p = spec["proc"]
                hb = spec["heartbeat"]
                stale = hb is not None and hb.value > 0 and (now - hb.value) > HEARTBEAT_TIMEOUT_S
                dead = not p.is_alive()
                if dead or stale:
                    print(f"[supervisor] {name}: dead={dead} stale={stale} → restart")
                    if p.is_alive():
                        p.terminate()
                        p.join(5.0)
                    if hb is not None:
                        hb.value = time.time()
                    fresh = mp.Process(
                        target=spec["target"], args=spec["args"], daemon=True, name=name
                    )
                    fresh.start()  # start BEFORE publishing (shutdown may join it)
                    spec["proc"] = fresh

    def shutdown(self):
        self.stop_event.set()
        for spec in self.specs.values():
            try:
                if spec["proc"].is_alive():
                    spec["proc"].terminate()
            except (AssertionError, ValueError):
                pass
        for spec in self.specs.values():
            try:
                spec["proc"].join(5.0)
                if spec["proc"].is_alive():
                    spec["proc"].kill()
            except (AssertionError, ValueError):
                pass  # a respawn raced shutdown; daemon procs die with us


# --------------------------------------------------------------------------- #
# Roles
# --------------------------------------------------------------------------- #
@register
def learner_sub_process(learner_ip, learner_port, *_):
    from pdrl_amd.agents.learner_module import is_on_policy
    from pdrl_amd.buffers import SharedRolloutRing, rollout_fields

    learner_port = int(learner_port)
    params = probe_env_spaces(Params)
    refresh_result_dirs(params)

    fields = rollout_fields(params.obs_dim, params.n_actions, params.hidden_size,
                            params.continuous)
    on_policy = is_on_policy(params.algo)
    capacity = params.batch_size if on_policy else params.buffer_size
    ring = SharedRolloutRing(fields, params.seq_len, capacity, on_policy)
    shared_stat = mp.Array("d", 3)
    stop_event = mp.Event()
    sup = Supervisor(stop_event)
    _install_cleanup(sup)

    # sharded ingest: each shard is its own decode+assemble process binding
    # its own data port; all shards write the one lock-guarded ring. The
    # manager routes rollouts across shards by worker connection.
    from pdrl_amd.agents import storage_shard_ports

    n_shards = max(1, int(getattr(params, "storage_shards", 1) or 1))
    for k, port in enumerate(storage_shard_ports(learner_port, n_shards)):
        hb_storage = mp.Value("d", time.time())
        sup.spawn(f"storage-{k}", storage_run,
                  (ring, learner_ip, port, params,
                   shared_stat if k == 0 else None, hb_storage, stop_event),
                  heartbeat=hb_storage)

    world_size = int(getattr(params, "num_learner_gpus", 1) or 1)
    if torch.cuda.is_available():
        world_size = min(world_size, torch.cuda.device_count())
    else:
        world_size = 1
    for rank in range(world_size):
        hb = mp.Value("d", time.time())
        sup.spawn(f"learner-{rank}", learner_run,
                  (ring, learner_ip, learner_port, params, shared_stat, hb, stop_event,
                   rank, world_size),
                  heartbeat=hb)
    sup.monitor()


@register
def manager_sub_process(manager_ip, learner_ip, port, learner_port, *_):
    stop_event = mp.Event()
    sup = Supervisor(stop_event)
    _install_cleanup(sup)
    n_shards = max(1, int(getattr(Params, "storage_shards", 1) or 1))
    hb = mp.Value("d", time.time())
    sup.spawn("manager", manager_run,
              (manager_ip, int(port), learner_ip, int(learner_port), hb, stop_event,
               n_shards),
              heartbeat=hb)
    sup.monitor()


@register
def worker_sub_process(num_p, manager_ip, learner_ip, port, learner_port, *_):
    params = probe_env_spaces(Params)
    model = build_model(params)
    load_actor_weights(model, params)
    # Each spawned worker gets its OWN pickled copy of this CPU model (no
    # share_memory: shared storages would make every worker's weight
    # hot-reload write the same tensors while siblings run inference —
    # transient torn weights mid-forward).
    model = model.cpu().eval()

    stop_event = mp.Event()
    sup = Supervisor(stop_event)
    _install_cleanup(sup)
    for i in range(int(num_p)):
        hb = mp.Value("d", time.time())
        sup.spawn(f"worker-{i}", worker_run,
                  (model, i, manager_ip, int(port), learner_ip, int(learner_port),
                   params, hb, stop_event),
                  heartbeat=hb)
    sup.monitor()


# --------------------------------------------------------------------------- #
def _install_cleanup(sup: Supervisor):
    def _handler(signum, frame):
        sup.shutdown()
        sys.exit(0)

    signal.signal(signal.SIGINT, _handler)
    signal.signal(signal.SIGTERM, _handler)
    atexit.register(sup.shutdown)


class Runner:
    def __init__(self):
        try:
            mp.set_start_method("spawn")
        except RuntimeError:
            pass

    def start(self):
        if len(sys.argv) < 2 or sys.argv[1] not in fn_dict:
            roles = ", ".join(fn_dict)
            print(f"usage: python main.py <{roles}> <args...>")
            sys.exit(2)
        role, args = sys.argv[1], sys.argv[2:]
        try:
            fn_dict[role](*args)
        except Exception:
            save_error_log(role, traceback.format_exc())
            raise


if __name__ == "__main__":
    Runner().start()
