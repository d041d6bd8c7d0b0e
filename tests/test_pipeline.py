"""Single-process integration test: worker → manager → storage → learner over
loopback transport with the real CartPole env + PPO (BASELINE.json
configs[0]: the plumbing config that runs without a GPU)."""
import threading
import time

import pytest
import torch

from pdrl_amd.agents import Learner, LearnerStorage, Manager, Worker
from pdrl_amd.agents.learner_module import is_on_policy
from pdrl_amd.buffers import SharedRolloutRing, rollout_fields
from pdrl_amd.networks import MlpLSTMSingle
from pdrl_amd.transport import Endpoint


def _free_port() -> int:
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _free_port_pair() -> int:
    """A port p such that p and p+1 are both bindable (data + weight planes)."""
    import socket

    for _ in range(64):
        s1 = socket.socket()
        s1.bind(("127.0.0.1", 0))
        p = s1.getsockname()[1]
        s2 = socket.socket()
        try:
            s2.bind(("127.0.0.1", p + 1))
        except OSError:
            continue
        finally:
            s1.close()
            s2.close()
        return p
    raise RuntimeError("no free port pair")


@pytest.fixture
def pipeline_params(params):
    params.env = "CartPole-v1"
    params.algo = "PPO"
    params.obs_dim, params.n_actions, params.continuous = 4, 2, False
    params.batch_size = 4
    params.seq_len = 5
    # everything runs as THREADS here (prod uses processes): give the
    # worker loops a real sleep so the manager/storage/learner threads
    # are scheduled — the optimized tick otherwise outpaces GIL handoffs
    params.worker_step_sleep = 0.002
    return params


def test_full_pipeline_cpu(pipeline_params):
    """2 workers + manager + storage + learner in threads: trajectories flow,
    the learner updates, weights broadcast back, checkpoints appear."""
    p = pipeline_params
    import multiprocessing

    stop = threading.Event()

    mgr_port = _free_port()
    lrn_port = _free_port_pair()  # learner weight plane binds lrn_port + 1
    time.sleep(0.05)

    fields = rollout_fields(p.obs_dim, p.n_actions, p.hidden_size, p.continuous)
    ring = SharedRolloutRing(fields, p.seq_len, p.batch_size, is_on_policy(p.algo))
    shared_stat = multiprocessing.Array("d", 3)

    manager = Manager("127.0.0.1", mgr_port, "127.0.0.1", lrn_port)
    storage = LearnerStorage(ring, "127.0.0.1", lrn_port, p, shared_stat=shared_stat)
    learner = Learner(ring, "127.0.0.1", lrn_port, p, device="cpu",
                      shared_stat=shared_stat)
    model = MlpLSTMSingle(p.obs_dim, p.n_actions, p.seq_len, p.hidden_size)
    workers = [
        Worker(model, i, "127.0.0.1", mgr_port, "127.0.0.1", lrn_port, p, seed=i)
        for i in range(2)
    ]

    threads = [
        threading.Thread(target=manager.run, daemon=True),
        threading.Thread(target=storage.run, daemon=True),
    ]
    for w in workers:
        w.stop_event = stop
        threads.append(threading.Thread(target=w.collect, daemon=True))
    manager.stop_event = stop
    storage.stop_event = stop
    for t in threads:
        t.start()

    # run 3 learner updates synchronously in this thread
    try:
        learner.run(max_updates=3)
    finally:
        stop.set()
    assert learner.updater.update_count >= 1

    # weight broadcast reached workers: worker SUB should have seen a Model msg
    time.sleep(0.5)
    # checkpoint written (model_save_interval=2)
    from pathlib import Path

    ckpts = list(Path(p.model_dir).glob("PPO_*.pt"))
    assert ckpts, f"no checkpoint in {p.model_dir}"
    # scalars logged
    from pdrl_amd.utils.logger import read_scalars

    scal = read_scalars(p.result_dir)
    assert "loss-total" in scal

    for w in workers:
        w.close()
    manager.close()
    storage.close()
    learner.close()
    for t in threads:
        t.join(2.0)


def test_worker_weight_hot_reload(pipeline_params):
    """A Model broadcast updates the worker's actor in place."""
    p = pipeline_params
    lrn_pub = Endpoint(bind=("127.0.0.1", 0))
    base = lrn_pub.bound_port - 1  # worker subscribes at learner_port + 1
    mgr_sub = Endpoint(bind=("127.0.0.1", 0))

    model = MlpLSTMSingle(p.obs_dim, p.n_actions, p.seq_len, p.hidden_size)
    w = Worker(model, 0, "127.0.0.1", mgr_sub.bound_port, "127.0.0.1", base, p, seed=0)
    assert lrn_pub.wait_peer(5.0)

    new_model = MlpLSTMSingle(p.obs_dim, p.n_actions, p.seq_len, p.hidden_size)
    from pdrl_amd.utils import Protocol, encode

    state = {k: v.cpu() for k, v in new_model.actor.state_dict().items()}
    lrn_pub.send(*encode(Protocol.Model, state))
    deadline = time.monotonic() + 5.0
    matched = False
    while time.monotonic() < deadline and not matched:
        w.poll_model()
        matched = all(
            torch.allclose(a, b)
            for a, b in zip(w.model.actor.state_dict().values(), state.values())
        )
        time.sleep(0.05)
    assert matched
    w.close()
    lrn_pub.close()
    mgr_sub.close()


def test_find_latest_checkpoint(tmp_path):
    from pdrl_amd.agents import find_latest_checkpoint

    d = tmp_path / "models"
    d.mkdir()
    for i in (1, 3, 20, 9):
        torch.save({}, d / f"IMPALA_{i}.pt")
    assert find_latest_checkpoint(str(d), "IMPALA").endswith("IMPALA_20.pt")
    assert find_latest_checkpoint(str(d), "PPO") is None
    assert find_latest_checkpoint(str(tmp_path / "nope"), "PPO") is None


def test_learner_resume_from_checkpoint(pipeline_params, tmp_path):
    """Learner(resume_path=...) restores updater state (reference:
    set_model_weight, main.py:128-146)."""
    import multiprocessing

    from pdrl_amd.agents import Learner
    from pdrl_amd.agents.learner_module import PPOUpdater
    from pdrl_amd.buffers import SharedRolloutRing, rollout_fields
    from pdrl_amd.networks import MlpLSTMSingle
    from tests.conftest import make_batch

    p = pipeline_params
    torch.manual_seed(3)
    model = MlpLSTMSingle(p.obs_dim, p.n_actions, p.seq_len, p.hidden_size)
    upd = PPOUpdater(model, p, "cpu")
    upd.step(make_batch(p))
    ckpt = tmp_path / "PPO_1.pt"
    upd.save(ckpt)

    fields = rollout_fields(p.obs_dim, p.n_actions, p.hidden_size, p.continuous)
    ring = SharedRolloutRing(fields, p.seq_len, p.batch_size, True)
    lrn_port = _free_port_pair()
    lrn = Learner(ring, "127.0.0.1", lrn_port, p, device="cpu",
                  shared_stat=multiprocessing.Array("d", 3),
                  resume_path=str(ckpt))
    assert lrn.updater.update_count == 1
    for p1, p2 in zip(model.parameters(),
                      lrn.updater.trainable_modules()["model"].parameters()):
        torch.testing.assert_close(p1.detach(), p2.detach())
    lrn.close()


def test_scalar_worker_path_still_works(pipeline_params, monkeypatch):
    """The per-step dict worker loop (PDRL_SCALAR_WORKER=1 debug path)
    keeps producing decodable rollout messages."""
    import torch

    from pdrl_amd.agents import Worker
    from pdrl_amd.networks import MlpLSTMSingle
    from pdrl_amd.transport import Endpoint

    monkeypatch.setenv("PDRL_SCALAR_WORKER", "1")
    p = pipeline_params
    model = MlpLSTMSingle(p.obs_dim, p.n_actions, p.seq_len, p.hidden_size)
    sub = Endpoint(bind=("127.0.0.1", 0))
    w = Worker(model, 0, "127.0.0.1", sub.bound_port, "127.0.0.1", 1, p,
               seed=0)
    w.collect(max_episodes=2)
    n = 0
    while sub.recv(timeout=0.3) is not None:
        n += 1
    w.close()
    sub.close()
    assert n >= 2  # at least per-episode flushes arrived
