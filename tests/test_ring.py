"""SharedRolloutRing tests: on-policy drain semantics, off-policy sampling,
cross-process producer/consumer correctness."""
import numpy as np
import torch
import torch.multiprocessing as mp

from pdrl_amd.buffers import SharedRolloutRing, rollout_fields


FIELDS = {"obs": 4, "rew": 1}


def make_traj(v, seq=5):
    return {
        "obs": torch.full((seq, 4), float(v)),
        "rew": torch.full((seq, 1), float(v)),
    }


def test_on_policy_drain_order_and_reset():
    ring = SharedRolloutRing(FIELDS, seq_len=5, capacity=4, on_policy=True)
    for v in range(4):
        assert ring.put(make_traj(v))
    assert ring.available() == 4
    batch = ring.drain_batch(4)
    np.testing.assert_array_equal(batch["rew"][:, 0, 0], [0, 1, 2, 3])
    assert ring.available() == 0
    # ring reusable after drain
    assert ring.put(make_traj(9))
    assert ring.available() == 1


def test_on_policy_full_drops():
    ring = SharedRolloutRing(FIELDS, seq_len=5, capacity=2, on_policy=True)
    assert ring.put(make_traj(0))
    assert ring.put(make_traj(1))
    assert not ring.put(make_traj(2))  # full → dropped, not overwritten
    batch = ring.drain_batch(2)
    np.testing.assert_array_equal(batch["rew"][:, 0, 0], [0, 1])


def test_off_policy_overwrite_and_sample():
    ring = SharedRolloutRing(FIELDS, seq_len=5, capacity=8, on_policy=False)
    rng = np.random.default_rng(0)
    assert ring.sample_batch(4, rng) is None  # not enough yet
    for v in range(20):  # wraps around: slots hold values 12..19
        ring.put(make_traj(v))
    batch = ring.sample_batch(8, rng)
    vals = set(batch["rew"][:, 0, 0].tolist())
    assert vals <= set(range(12, 20))


def test_rollout_fields_widths():
    f = rollout_fields(obs_dim=4, n_actions=2, hidden=64, continuous=False)
    assert f == {
        "obs": 4, "act": 1, "rew": 1, "logits": 2,
        "log_prob": 1, "is_fir": 1, "hx": 64, "cx": 64,
    }
    fc = rollout_fields(obs_dim=2, n_actions=1, hidden=64, continuous=True)
    assert fc["act"] == 1 and fc["logits"] == 2


def _producer(ring, n):
    for v in range(n):
        while not ring.put(make_traj(v)):
            pass


def test_cross_process_producer_consumer():
    ctx = mp.get_context("spawn")
    ring = SharedRolloutRing(FIELDS, seq_len=5, capacity=8, on_policy=True)
    p = ctx.Process(target=_producer, args=(ring, 24))
    p.start()
    seen = []
    while len(seen) < 24:
        b = ring.drain_batch(8)
        if b is not None:
            seen.extend(b["rew"][:, 0, 0].tolist())
    p.join(10)
    assert not p.is_alive()
    assert seen == [float(v) for v in range(24)]


def test_drain_new_off_policy():
    ring = SharedRolloutRing(FIELDS, seq_len=5, capacity=8, on_policy=False)
    assert ring.drain_new() is None
    for v in range(5):
        ring.put(make_traj(v))
    out = ring.drain_new()
    np.testing.assert_array_equal(out["rew"][:, 0, 0], [0, 1, 2, 3, 4])
    assert ring.drain_new() is None  # nothing new
    for v in range(5, 20):  # overwrites: only last 8 retrievable
        ring.put(make_traj(v))
    out = ring.drain_new()
    np.testing.assert_array_equal(out["rew"][:, 0, 0], list(range(12, 20)))


def test_drain_new_handles_producer_lap():
    """Off-policy conveyor: if the producer laps the consumer, drain_new
    resumes from the oldest still-present slot instead of re-reading
    overwritten data."""
    import numpy as np
    import torch

    from pdrl_amd.buffers import SharedRolloutRing

    ring = SharedRolloutRing({"x": 2}, seq_len=3, capacity=4, on_policy=False)
    def traj(v):
        return {"x": torch.full((3, 2), float(v))}

    for v in range(10):  # laps the 4-slot ring twice without any drain
        ring.put(traj(v))
    out = ring.drain_new()
    # only the newest capacity=4 trajectories (6, 7, 8, 9) survive
    assert out["x"].shape == (4, 3, 2)
    np.testing.assert_allclose(out["x"][:, 0, 0], [6.0, 7.0, 8.0, 9.0])
    # nothing new since
    assert ring.drain_new() is None
    ring.put(traj(10))
    out = ring.drain_new()
    np.testing.assert_allclose(out["x"][:, 0, 0], [10.0])
