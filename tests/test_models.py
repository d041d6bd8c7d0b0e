"""Model zoo tests: SeqLSTMCore parity vs torch.nn.LSTMCell, shapes of all
13 model classes, act/forward contracts."""
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from pdrl_amd.networks import (
    MlpLSTMActor,
    MlpLSTMActorContinuous,
    MlpLSTMBase,
    MlpLSTMContinuous,
    MlpLSTMCritic,
    MlpLSTMCriticContinuous,
    MlpLSTMDoubleCritic,
    MlpLSTMDoubleCriticContinuous,
    MlpLSTMSeperate,
    MlpLSTMSeperateContinuous,
    MlpLSTMSingle,
    MlpLSTMSingleContinuous,
    SeqLSTMCore,
    categorical_stats,
)

B, S, F_DIM, H, A = 6, 5, 4, 64, 2


def test_core_matches_nn_lstmcell():
    """The explicit gate math must equal torch's LSTMCell given the same
    weights (this is the oracle the HIP kernel is later tested against)."""
    torch.manual_seed(0)
    core = SeqLSTMCore(F_DIM, H, {"out": 3})
    cell = nn.LSTMCell(H, H)
    with torch.no_grad():
        cell.weight_ih.copy_(core.w_ih.t())
        cell.weight_hh.copy_(core.w_hh.t())
        cell.bias_ih.copy_(core.b_g)
        cell.bias_hh.zero_()

    x = torch.randn(B, S, F_DIM)
    hx, cx = torch.randn(B, H) * 0.1, torch.randn(B, H) * 0.1
    outs, h_end, c_end = core(x, hx, cx)

    xb = F.relu(x.reshape(B * S, F_DIM) @ core.body_w + core.body_b).view(B, S, H)
    h, c = hx, cx
    for t in range(S):
        h, c = cell(xb[:, t], (h, c))
    torch.testing.assert_close(h_end, h, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(c_end, c, rtol=1e-5, atol=1e-5)
    w, b = core.head_params("out")
    torch.testing.assert_close(outs["out"][:, -1], h @ w + b, rtol=1e-5, atol=1e-5)


def test_core_backward_flows():
    core = SeqLSTMCore(F_DIM, H, {"v": 1})
    x = torch.randn(B, S, F_DIM, requires_grad=True)
    outs, _, _ = core(x, torch.zeros(B, H), torch.zeros(B, H))
    outs["v"].sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    for p in core.parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all()


def test_base_act_and_forward_shapes():
    m = MlpLSTMBase(F_DIM, A, S, H)
    obs = torch.randn(1, F_DIM)
    action, logits, log_prob, (hx, cx) = m.act(obs, (torch.zeros(1, H), torch.zeros(1, H)))
    assert action.shape == (1, 1) and logits.shape == (1, A)
    assert log_prob.shape == (1, 1) and hx.shape == (1, H)
    assert action.item() in (0, 1)

    obs_b = torch.randn(B, S, F_DIM)
    acts = torch.randint(0, A, (B, S, 1)).float()
    lg, lp, ent, val = m(obs_b, (torch.zeros(B, H), torch.zeros(B, H)), acts)
    assert lg.shape == (B, S, A) and lp.shape == (B, S, 1)
    assert ent.shape == (B, S, 1) and val.shape == (B, S, 1)
    assert (ent >= 0).all()


def test_categorical_stats_match_distributions():
    torch.manual_seed(1)
    logits = torch.randn(B, S, A)
    acts = torch.randint(0, A, (B, S, 1))
    lp, ent = categorical_stats(logits, acts)
    d = torch.distributions.Categorical(logits=logits)
    torch.testing.assert_close(lp.squeeze(-1), d.log_prob(acts.squeeze(-1)), rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(ent.squeeze(-1), d.entropy(), rtol=1e-5, atol=1e-6)


def test_continuous_models():
    m = MlpLSTMContinuous(2, 1, S, H)
    a, logits, lp, (hx, cx) = m.act(torch.randn(1, 2), (torch.zeros(1, H), torch.zeros(1, H)))
    assert a.shape == (1, 1) and logits.shape == (1, 2)
    lg, lp2, ent, val = m(torch.randn(B, S, 2), (torch.zeros(B, H), torch.zeros(B, H)),
                          torch.randn(B, S, 1))
    assert lp2.shape == (B, S, 1) and val.shape == (B, S, 1)


def test_sac_discrete_stack():
    m = MlpLSTMSeperate(F_DIM, A, S, H)
    probs, log_probs = m.actor(torch.randn(B, S, F_DIM), (torch.zeros(B, H), torch.zeros(B, H)))
    assert probs.shape == (B, S, A)
    torch.testing.assert_close(probs.sum(-1), torch.ones(B, S), rtol=1e-5, atol=1e-5)
    q1, q2 = m.critic(torch.randn(B, S, F_DIM), (torch.zeros(B, H), torch.zeros(B, H)))
    assert q1.shape == (B, S, A)
    # twin critics are independent networks
    assert not torch.allclose(q1, q2)


def test_sac_continuous_stack():
    m = MlpLSTMSeperateContinuous(2, 1, S, H)
    act, lp = m.actor(torch.randn(B, S, 2), (torch.zeros(B, H), torch.zeros(B, H)))
    assert act.shape == (B, S, 1) and (act.abs() <= 1).all()
    assert lp.shape == (B, S, 1)
    q1, q2 = m.critic(torch.randn(B, S, 2), act, (torch.zeros(B, H), torch.zeros(B, H)))
    assert q1.shape == (B, S, 1)
    # reparameterized action carries gradient
    act.sum().backward()
    assert any(p.grad is not None for p in m.actor.parameters())


def test_single_wrapper_aliases_critic():
    m = MlpLSTMSingle(F_DIM, A, S, H)
    assert m.critic is m.actor  # single-network algos share the torso


def test_all_13_classes_instantiate():
    classes = [
        MlpLSTMBase, MlpLSTMContinuous, MlpLSTMActor, MlpLSTMActorContinuous,
        MlpLSTMCritic, MlpLSTMCriticContinuous, MlpLSTMDoubleCritic,
        MlpLSTMDoubleCriticContinuous, MlpLSTMSingle, MlpLSTMSingleContinuous,
        MlpLSTMSeperate, MlpLSTMSeperateContinuous, SeqLSTMCore,
    ]
    for cls in classes:
        if cls is SeqLSTMCore:
            cls(F_DIM, H, {"o": 1})
        else:
            cls(F_DIM, A, S, H)


def test_cpu_actor_matches_eager_act():
    """C++ batched actor (ops/csrc/cpu_actor.cpp) vs the eager core: the
    deterministic pieces (logits, h, c) match to fp32 tolerance; the
    sampled action's log-prob equals log_softmax(logits)[action]; the
    sampled action distribution tracks the policy."""
    pytest.importorskip("pdrl_amd.ops._cpu_actor")
    import torch
    import torch.nn.functional as Fn

    from pdrl_amd.networks import MlpLSTMSingle
    from pdrl_amd.ops import _cpu_actor

    torch.manual_seed(3)
    model = MlpLSTMSingle(4, 3, 5, 64).eval()
    core = model.actor.core
    M = 6
    obs = torch.randn(M, 4)
    hx = torch.randn(M, 64) * 0.2
    cx = torch.randn(M, 64) * 0.2
    rng = torch.tensor([12345], dtype=torch.int64)

    a, lg, lp, h, c = _cpu_actor.act_batch_discrete(
        obs, hx, cx, core.body_w.detach(), core.body_b.detach(),
        core.w_ih.detach(), core.w_hh.detach(), core.b_g.detach(),
        core.heads_w.detach(), core.heads_b.detach(), 3, rng)

    outs, he, ce = core.step(obs, hx, cx)
    torch.testing.assert_close(lg, outs["logits"], rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(h, he, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(c, ce, rtol=1e-4, atol=1e-5)

    ref_lp = Fn.log_softmax(outs["logits"], dim=-1).gather(-1, a)
    torch.testing.assert_close(lp, ref_lp, rtol=1e-4, atol=1e-5)
    assert a.min() >= 0 and a.max() < 3

    # empirical action frequencies track the softmax policy (chi-ish check)
    probs = Fn.softmax(outs["logits"][0:1], dim=-1)
    counts = torch.zeros(3)
    for _ in range(2000):
        a1, *_ = _cpu_actor.act_batch_discrete(
            obs[0:1], hx[0:1], cx[0:1], core.body_w.detach(),
            core.body_b.detach(), core.w_ih.detach(), core.w_hh.detach(),
            core.b_g.detach(), core.heads_w.detach(), core.heads_b.detach(),
            3, rng)
        counts[a1.item()] += 1
    freq = counts / counts.sum()
    assert (freq - probs.squeeze(0)).abs().max() < 0.05, (freq, probs)


def test_worker_fast_act_engages_and_rolls():
    """Worker picks the C++ act path for discrete policies and produces
    well-formed records end to end (FakeEnv, no transport)."""
    pytest.importorskip("pdrl_amd.ops._cpu_actor")
    import torch

    from pdrl_amd.agents.worker import Worker
    from pdrl_amd.networks import MlpLSTMSingle
    from pdrl_amd.utils import load_params

    p = load_params()
    p.env = "FakeEnv"
    p.algo = "IMPALA"
    p.continuous = False
    p.num_envs_per_worker = 2

    torch.manual_seed(0)
    model = MlpLSTMSingle(4, 2, p.seq_len, p.hidden_size)

    sent = []

    class _StubPub:
        def send(self, h, b):
            sent.append((h, b))

        def close(self):
            pass

    w = Worker.__new__(Worker)
    w.params = p
    w.worker_idx = 0
    w.model = model.eval()
    w._continuous = False
    w._act = None
    act = w._make_fast_act(seed=1)
    assert act is not None, "fast act must engage for the discrete actor"
    obs = torch.randn(2, 4)
    hx = torch.zeros(2, p.hidden_size)
    cx = torch.zeros(2, p.hidden_size)
    a, lg, lp, (h, c) = act(obs, (hx, cx))
    assert a.shape == (2, 1) and lg.shape == (2, 2) and lp.shape == (2, 1)
    assert h.shape == (2, p.hidden_size)


def test_cpu_actor_gaussian_matches_eager():
    """C++ Gaussian act (both sampling schemes) vs the eager actors: the
    deterministic pieces (logits record, h, c) match; sampled actions and
    log-probs are mutually consistent under the recorded distribution."""
    pytest.importorskip("pdrl_amd.ops._cpu_actor")
    import math

    import torch

    from pdrl_amd.networks import MlpLSTMActorContinuous, MlpLSTMContinuous
    from pdrl_amd.ops import _cpu_actor

    torch.manual_seed(5)
    M, A = 5, 2

    for mode, cls in ((0, MlpLSTMContinuous), (1, MlpLSTMActorContinuous)):
        model = cls(3, A, 5, 64).eval()
        core = model.core
        obs = torch.randn(M, 3)
        hx = torch.randn(M, 64) * 0.1
        cx = torch.randn(M, 64) * 0.1
        rng = torch.tensor([777], dtype=torch.int64)
        a, lg, lp, h, c = _cpu_actor.act_batch_gaussian(
            obs, hx, cx, core.body_w.detach(), core.body_b.detach(),
            core.w_ih.detach(), core.w_hh.detach(), core.b_g.detach(),
            core.heads_w.detach(), core.heads_b.detach(), A, mode, rng)

        outs, he, ce = core.step(obs, hx, cx)
        torch.testing.assert_close(h, he, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(c, ce, rtol=1e-4, atol=1e-5)
        second = outs["std"] if mode == 0 else outs["log_std"]
        torch.testing.assert_close(
            lg, torch.cat([outs["mu"], second], -1), rtol=1e-4, atol=1e-5)

        # log-prob consistency with the recorded (mu, second) under the
        # scheme's own density
        mu, sec = lg[..., :A], lg[..., A:]
        if mode == 0:
            mean = torch.tanh(mu)
            sd = torch.nn.functional.softplus(sec) + 1e-4
            ref = (-0.5 * ((a - mean) / sd) ** 2 - sd.log()
                   - 0.5 * math.log(2 * math.pi)).sum(-1, keepdim=True)
        else:
            ls = sec.clamp(-20.0, 2.0)
            sd = ls.exp()
            z = torch.atanh(a.clamp(-1 + 1e-6, 1 - 1e-6))
            ref = (-0.5 * ((z - mu) / sd) ** 2 - ls
                   - 0.5 * math.log(2 * math.pi)
                   - torch.log(1 - a.pow(2) + 1e-7)).sum(-1, keepdim=True)
            assert a.abs().max() < 1.0  # tanh-squashed
        torch.testing.assert_close(lp, ref, rtol=1e-3, atol=1e-4)
