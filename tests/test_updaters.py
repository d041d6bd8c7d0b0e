"""Algorithm updater tests: each of the 5 algorithms runs a step on a
synthetic batch, produces finite losses, and changes parameters."""
import copy

import numpy as np

import pytest
import torch

from pdrl_amd.agents.learner_module import (
    ImpalaUpdater,
    PPOUpdater,
    SACContinuousUpdater,
    SACUpdater,
    VMPOUpdater,
    is_on_policy,
    switch_module,
)
from pdrl_amd.networks import (
    MlpLSTMSeperate,
    MlpLSTMSeperateContinuous,
    MlpLSTMSingle,
    MlpLSTMSingleContinuous,
)
from tests.conftest import make_batch


def param_snapshot(module):
    return [p.detach().clone() for p in module.parameters()]


def params_changed(before, module):
    return any(not torch.allclose(b, p.detach()) for b, p in zip(before, module.parameters()))


@pytest.mark.parametrize("updater_cls,model_cls", [
    (PPOUpdater, MlpLSTMSingle),
    (ImpalaUpdater, MlpLSTMSingle),
    (VMPOUpdater, MlpLSTMSingle),
])
def test_on_policy_updaters_step(updater_cls, model_cls, params):
    torch.manual_seed(0)
    params.obs_dim, params.n_actions = 4, 2
    model = model_cls(4, 2, params.seq_len, params.hidden_size)
    upd = updater_cls(model, params, "cpu")
    batch = make_batch(params)
    before = param_snapshot(model)
    stats = upd.step(batch)
    assert all(np.isfinite(float(v)) for v in stats.values()), stats
    assert params_changed(before, model)
    assert upd.update_count == 1


def test_ppo_continuous(params):
    torch.manual_seed(0)
    params.obs_dim, params.n_actions = 2, 1
    params.algo = "PPO-Continuous"
    model = MlpLSTMSingleContinuous(2, 1, params.seq_len, params.hidden_size)
    upd = PPOUpdater(model, params, "cpu")
    batch = make_batch(params, n_actions=1, continuous=True)
    batch["obs"] = torch.randn(params.batch_size, params.seq_len, 2)
    stats = upd.step(batch)
    assert all(np.isfinite(float(v)) for v in stats.values())


def test_sac_discrete(params):
    torch.manual_seed(0)
    params.obs_dim, params.n_actions = 4, 2
    model = MlpLSTMSeperate(4, 2, params.seq_len, params.hidden_size)
    upd = SACUpdater(model, params, "cpu")
    # target critic is a REAL deep copy
    assert upd.target_critic is not upd.critic
    for p, tp in zip(upd.critic.parameters(), upd.target_critic.parameters()):
        assert p is not tp
        torch.testing.assert_close(p, tp)
    batch = make_batch(params)
    before_t = param_snapshot(upd.target_critic)
    stats = upd.step(batch)
    assert all(np.isfinite(float(v)) for v in stats.values())
    # soft update moved the target
    assert params_changed(before_t, upd.target_critic)


def test_sac_continuous(params):
    torch.manual_seed(0)
    params.obs_dim, params.n_actions = 2, 1
    model = MlpLSTMSeperateContinuous(2, 1, params.seq_len, params.hidden_size)
    upd = SACContinuousUpdater(model, params, "cpu")
    batch = make_batch(params, n_actions=1, continuous=True)
    batch["obs"] = torch.randn(params.batch_size, params.seq_len, 2)
    alpha_before = float(upd.log_alpha)
    stats = upd.step(batch)
    assert all(np.isfinite(float(v)) for v in stats.values())
    assert float(upd.log_alpha) != alpha_before  # temperature auto-tuned


def test_vmpo_duals_update(params):
    torch.manual_seed(0)
    params.obs_dim, params.n_actions = 4, 2
    model = MlpLSTMSingle(4, 2, params.seq_len, params.hidden_size)
    upd = VMPOUpdater(model, params, "cpu")
    eta0, alpha0 = float(upd.log_eta), float(upd.log_alpha)
    upd.step(make_batch(params))
    assert float(upd.log_eta) != eta0 or float(upd.log_alpha) != alpha0


def test_module_switcher():
    for algo in ("PPO", "IMPALA", "V-MPO", "SAC", "SAC-Continuous", "PPO-Continuous"):
        upd, mdl = switch_module(algo)
        assert upd is not None and mdl is not None
    assert is_on_policy("PPO") and is_on_policy("IMPALA") and is_on_policy("V-MPO")
    assert not is_on_policy("SAC") and not is_on_policy("SAC-Continuous")
    with pytest.raises(ValueError):
        switch_module("DQN")


def test_checkpoint_roundtrip(params, tmp_path):
    torch.manual_seed(0)
    params.obs_dim, params.n_actions = 4, 2
    model = MlpLSTMSingle(4, 2, params.seq_len, params.hidden_size)
    upd = PPOUpdater(model, params, "cpu")
    upd.step(make_batch(params))
    path = tmp_path / "PPO_1.pt"
    upd.save(path)

    model2 = MlpLSTMSingle(4, 2, params.seq_len, params.hidden_size)
    upd2 = PPOUpdater(model2, params, "cpu")
    upd2.load(path)
    assert upd2.update_count == 1
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        torch.testing.assert_close(p1, p2)


def test_checkpoint_persists_loose_duals(params, tmp_path):
    """SAC's log_alpha and V-MPO's log_eta/log_alpha live outside any module;
    save/load must round-trip them or resume silently resets the temperature
    and dual variables."""
    from pdrl_amd.agents.learner_module import SACUpdater, VMPOUpdater
    from pdrl_amd.networks import MlpLSTMSeperate, MlpLSTMSingle

    torch.manual_seed(1)
    params.obs_dim, params.n_actions = 4, 2

    sac = SACUpdater(MlpLSTMSeperate(4, 2, params.seq_len, params.hidden_size),
                     params, "cpu")
    with torch.no_grad():
        sac.log_alpha.fill_(-1.2345)
    sac.save(tmp_path / "SAC_1.pt")
    sac2 = SACUpdater(MlpLSTMSeperate(4, 2, params.seq_len, params.hidden_size),
                      params, "cpu")
    sac2.load(tmp_path / "SAC_1.pt")
    assert float(sac2.log_alpha) == pytest.approx(-1.2345)

    vmpo = VMPOUpdater(MlpLSTMSingle(4, 2, params.seq_len, params.hidden_size),
                       params, "cpu")
    with torch.no_grad():
        vmpo.log_eta.fill_(0.777)
        vmpo.log_alpha.fill_(-0.333)
    vmpo.save(tmp_path / "V-MPO_1.pt")
    vmpo2 = VMPOUpdater(MlpLSTMSingle(4, 2, params.seq_len, params.hidden_size),
                        params, "cpu")
    vmpo2.load(tmp_path / "V-MPO_1.pt")
    assert float(vmpo2.log_eta) == pytest.approx(0.777)
    assert float(vmpo2.log_alpha) == pytest.approx(-0.333)


@pytest.mark.parametrize("B,S", [(3, 3), (7, 9), (1, 5)])
@pytest.mark.parametrize("algo", ["PPO", "IMPALA", "V-MPO", "SAC"])
def test_updaters_odd_shapes(algo, B, S, params):
    """Non-default batch/sequence shapes step without shape assumptions
    leaking (the GPU fused paths gate on shape; the CPU reference path
    must be shape-agnostic)."""
    from pdrl_amd.agents.learner_module import switch_module

    torch.manual_seed(1)
    params.algo = algo
    params.batch_size, params.seq_len = B, S
    params.obs_dim, params.n_actions = 4, 2
    upd_cls, model_cls = switch_module(algo)
    model = model_cls(4, 2, S, params.hidden_size)
    upd = upd_cls(model, params, "cpu")
    batch = make_batch(params)
    stats = upd.step(batch)
    assert all(np.isfinite(float(v)) for v in stats.values()), (algo, B, S)
