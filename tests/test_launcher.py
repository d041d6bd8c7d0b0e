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


def test_refresh_result_dirs_respects_explicit_config(tmp_path):
    """An explicitly configured result_dir/model_dir survives
    refresh_result_dirs, making checkpoint resume reachable (a fresh
    timestamped dir used to clobber it before resume could scan it)."""
    import json

    from pdrl_amd.utils import load_params, refresh_result_dirs

    cfg = json.loads((
        __import__("pathlib").Path("pdrl_amd/utils/parameters.json")
    ).read_text())
    cfg["result_dir"] = str(tmp_path / "runA")
    cfg["model_dir"] = str(tmp_path / "runA" / "models")
    f = tmp_path / "params.json"
    f.write_text(json.dumps(cfg))

    p = load_params(f)
    assert p.explicit_dirs
    before = (p.result_dir, p.model_dir)
    refresh_result_dirs(p)
    assert (p.result_dir, p.model_dir) == before

    # default (null dirs) config still gets a fresh stamp
    cfg["result_dir"] = None
    cfg["model_dir"] = None
    f.write_text(json.dumps(cfg))
    q = load_params(f)
    assert not q.explicit_dirs
    old = q.result_dir
    import time as _t
    _t.sleep(1.1)
    refresh_result_dirs(q)
    assert q.result_dir != old


def test_learner_run_resolves_resume_at_start(tmp_path, monkeypatch):
    """learner_run finds the newest checkpoint AT (re)start — a Supervisor
    respawn resumes saved progress instead of re-initializing (the round-1
    design froze a None resume_path into the spawn args)."""
    import torch

    import main as main_mod
    from pdrl_amd.utils import load_params

    p = load_params()
    p.algo = "IMPALA"
    p.obs_dim, p.n_actions, p.continuous = 4, 2, False
    p.model_dir = str(tmp_path / "models")
    (tmp_path / "models").mkdir()

    from pdrl_amd.agents.learner_module import switch_module

    upd_cls, model_cls = switch_module("IMPALA")
    torch.manual_seed(7)
    model = model_cls(4, 2, p.seq_len, p.hidden_size)
    upd = upd_cls(model, p, "cpu")
    upd.update_count = 30
    upd.save(tmp_path / "models" / "IMPALA_30.pt")

    captured = {}

    class _FakeLearner:
        def __init__(self, *a, **kw):
            captured["resume_path"] = kw.get("resume_path")
            self.updater = upd

        def run(self):
            pass

    monkeypatch.setattr("pdrl_amd.agents.Learner", _FakeLearner)
    main_mod.learner_run(None, "127.0.0.1", 40123, p, None, None, None, 0, 1)
    assert captured["resume_path"].endswith("IMPALA_30.pt")
