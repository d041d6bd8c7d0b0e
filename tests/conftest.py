import os
import sys
from pathlib import Path

import pytest
import torch

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a (MI355X) GPU; run with -m gpu")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def params(tmp_path):
    """Small, fast hyperparameter namespace for tests."""
    from pdrl_amd.utils import load_params

    p = load_params()
    p.seq_len = 5
    p.batch_size = 8
    p.hidden_size = 64
    p.time_horizon = 64
    p.K_epoch = 1
    p.loss_log_interval = 1
    p.model_save_interval = 2
    p.result_dir = str(tmp_path / "results")
    p.model_dir = str(tmp_path / "results" / "models")
    p.worker_step_sleep = 0.0
    return p


def make_batch(params, n_actions=2, continuous=False, device="cpu", seed=0):
    """Random synthetic trajectory batch shaped like the shared ring output."""
    g = torch.Generator().manual_seed(seed)
    B, S, H = params.batch_size, params.seq_len, params.hidden_size
    obs_dim = getattr(params, "obs_dim", 4)
    act_dim = n_actions if continuous else 1
    logits_dim = 2 * n_actions if continuous else n_actions
    batch = {
        "obs": torch.randn(B, S, obs_dim, generator=g),
        "rew": torch.rand(B, S, 1, generator=g),
        "logits": torch.randn(B, S, logits_dim, generator=g),
        "log_prob": -torch.rand(B, S, 1, generator=g),
        "is_fir": (torch.rand(B, S, 1, generator=g) < 0.1).float(),
        "hx": torch.randn(B, S, H, generator=g) * 0.1,
        "cx": torch.randn(B, S, H, generator=g) * 0.1,
    }
    if continuous:
        batch["act"] = torch.tanh(torch.randn(B, S, act_dim, generator=g))
    else:
        batch["act"] = torch.randint(0, n_actions, (B, S, 1), generator=g).float()
    return {k: v.to(device) for k, v in batch.items()}
