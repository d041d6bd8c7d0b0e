"""DeviceReplay: CPU-device unit tests (logic) + GPU test (HBM residency)."""
import numpy as np
import pytest
import torch

from pdrl_amd.buffers.device_replay import DeviceReplay

FIELDS = {"obs": 4, "rew": 1}


def make(n, start=0, seq=5):
    return {
        "obs": np.stack([np.full((seq, 4), float(v), np.float32) for v in range(start, start + n)]),
        "rew": np.stack([np.full((seq, 1), float(v), np.float32) for v in range(start, start + n)]),
    }


def test_append_wrap_and_sample_cpu():
    rep = DeviceReplay(FIELDS, seq_len=5, capacity=8, device="cpu")
    assert rep.sample(4) is None
    rep.append_batch(make(5))
    assert rep.size == 5
    rep.append_batch(make(6, start=5))  # wraps; holds 3..10
    assert rep.size == 8
    s = rep.sample(8)
    vals = set(s["rew"][:, 0, 0].tolist())
    assert vals <= set(float(v) for v in range(3, 11))
    assert s["obs"].shape == (8, 5, 4)


def test_capacity_accounting():
    rep = DeviceReplay(FIELDS, seq_len=5, capacity=4, device="cpu")
    rep.append_batch(make(4))
    rep.append_batch(make(4, start=100))
    s = rep.sample(4)
    assert set(s["rew"][:, 0, 0].tolist()) <= {100.0, 101.0, 102.0, 103.0}
    assert rep.nbytes() == 4 * 5 * 5 * 4


@pytest.mark.gpu
def test_device_replay_on_hbm():
    rep = DeviceReplay(FIELDS, seq_len=5, capacity=1024, device="cuda:0")
    rep.append_batch(make(100))
    s = rep.sample(32)
    assert s["obs"].is_cuda and s["rew"].is_cuda
    # sampling is device-side: no sync needed for correctness of devices
    assert s["obs"].shape == (32, 5, 4)
