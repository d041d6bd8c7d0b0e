"""Cluster launcher (run.py) and runner (main.py) CLI tests."""
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_run_dry_run_builds_commands():
    out = subprocess.run(
        [sys.executable, str(REPO / "run.py"), "--dry-run"],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr
    text = out.stdout
    assert "tmux new-session" in text
    assert "learner_sub_process" in text
    assert "manager_sub_process" in text
    assert "worker_sub_process" in text


def test_main_usage_on_bad_role():
    out = subprocess.run(
        [sys.executable, str(REPO / "main.py"), "not_a_role"],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 2
    assert "usage" in out.stdout


def test_probe_env_spaces():
    from pdrl_amd.utils import load_params
    import main as main_mod

    p = load_params()
    p.env, p.algo = "CartPole-v1", "PPO"
    main_mod.probe_env_spaces(p)
    assert p.obs_dim == 4 and p.n_actions == 2 and not p.continuous

    p.env, p.algo = "MountainCarContinuous-v0", "SAC-Continuous"
    main_mod.probe_env_spaces(p)
    assert p.obs_dim == 2 and p.n_actions == 1 and p.continuous


def test_probe_env_algo_mismatch():
    import pytest

    from pdrl_amd.utils import load_params
    import main as main_mod

    p = load_params()
    p.env, p.algo = "MountainCarContinuous-v0", "PPO"
    with pytest.raises(AssertionError):
        main_mod.probe_env_spaces(p)


def test_build_model_for_each_algo():
    from pdrl_amd.utils import load_params
    import main as main_mod

    for env, algo in [
        ("CartPole-v1", "PPO"),
        ("CartPole-v1", "IMPALA"),
        ("CartPole-v1", "V-MPO"),
        ("CartPole-v1", "SAC"),
        ("MountainCarContinuous-v0", "SAC-Continuous"),
        ("MountainCarContinuous-v0", "PPO-Continuous"),
    ]:
        p = load_params()
        p.env, p.algo = env, algo
        main_mod.probe_env_spaces(p)
        model = main_mod.build_model(p)
        assert model is not None
