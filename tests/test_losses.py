"""Loss-math tests: GAE / V-trace scans vs brute-force references,
soft_update, categorical KL."""
import copy

import torch

from pdrl_amd.agents.learner_module.compute_loss import (
    compute_gae,
    compute_v_trace,
    kl_divergence,
    soft_update,
)
from pdrl_amd.networks import MlpLSTMCritic

B, T = 4, 7


def brute_gae(deltas, gamma, lmbda, dones):
    B, T, _ = deltas.shape
    out = torch.zeros_like(deltas)
    for b in range(B):
        run = 0.0
        for t in reversed(range(T)):
            run = float(deltas[b, t]) + gamma * lmbda * (1 - float(dones[b, t])) * run
            out[b, t] = run
    return out


def test_gae_matches_bruteforce():
    torch.manual_seed(0)
    deltas = torch.randn(B, T, 1)
    dones = (torch.rand(B, T, 1) < 0.3).float()
    got = compute_gae(deltas, 0.99, 0.95, dones)
    want = brute_gae(deltas, 0.99, 0.95, dones)
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)


def test_gae_no_dones_geometric():
    deltas = torch.ones(1, 4, 1)
    got = compute_gae(deltas, 0.5, 1.0, None).squeeze()
    # adv_t = sum_{k>=t} 0.5^(k-t)
    want = torch.tensor([1.875, 1.75, 1.5, 1.0])
    torch.testing.assert_close(got, want)


def brute_vtrace(behav_lp, target_lp, is_fir, rew, val, gamma, rho_bar=0.8, rho_min=0.1, c_bar=1.0):
    B, S, _ = val.shape
    T = S - 1
    rhos = torch.clamp((target_lp - behav_lp)[:, :-1].exp(), rho_min, rho_bar)
    cs = torch.clamp((target_lp - behav_lp)[:, :-1].exp(), max=c_bar)
    mask = 1.0 - is_fir[:, 1:]
    vs = torch.zeros(B, T, 1)
    for b in range(B):
        acc = 0.0
        accs = [0.0] * T
        for t in reversed(range(T)):
            d = float(rhos[b, t]) * (
                float(rew[b, t]) + gamma * float(mask[b, t]) * float(val[b, t + 1]) - float(val[b, t])
            )
            acc = d + gamma * float(mask[b, t]) * float(cs[b, t]) * acc
            accs[t] = acc
        for t in range(T):
            vs[b, t] = val[b, t] + accs[t]
    adv = torch.zeros(B, T, 1)
    for b in range(B):
        for t in range(T):
            nxt = float(vs[b, t + 1]) if t + 1 < T else float(val[b, T])
            adv[b, t] = float(rhos[b, t]) * (
                float(rew[b, t]) + gamma * float(mask[b, t]) * nxt - float(val[b, t])
            )
    return rhos, adv, vs


def test_vtrace_matches_bruteforce():
    torch.manual_seed(1)
    S = T + 1
    behav = -torch.rand(B, S, 1)
    target = behav + 0.3 * torch.randn(B, S, 1)
    is_fir = (torch.rand(B, S, 1) < 0.2).float()
    rew = torch.randn(B, S, 1)
    val = torch.randn(B, S, 1)
    rhos, adv, vs = compute_v_trace(behav, target, is_fir, rew, val, 0.99)
    b_rhos, b_adv, b_vs = brute_vtrace(behav, target, is_fir, rew, val, 0.99)
    torch.testing.assert_close(rhos, b_rhos, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(vs, b_vs, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(adv, b_adv, rtol=1e-4, atol=1e-5)


def test_vtrace_on_policy_reduces_to_gae_lambda1():
    """With target==behaviour (rho=c=1 clamped to [0.1,0.8]→1 requires
    rho_bar≥1; use wide clamps) V-trace vs = TD(λ=1) returns."""
    torch.manual_seed(2)
    S = 6
    lp = -torch.rand(B, S, 1)
    is_fir = torch.zeros(B, S, 1)
    rew = torch.randn(B, S, 1)
    val = torch.randn(B, S, 1)
    rhos, adv, vs = compute_v_trace(lp, lp, is_fir, rew, val, 0.99, rho_bar=1.0, rho_min=0.0,
                                    c_bar=1.0)
    # vs_t should equal discounted return bootstrapped from val[:, -1]
    want = torch.zeros(B, S - 1, 1)
    boot = val[:, -1]
    for b in range(B):
        acc = float(boot[b])
        for t in reversed(range(S - 1)):
            acc = float(rew[b, t]) + 0.99 * acc
            want[b, t] = acc
    torch.testing.assert_close(vs, want, rtol=1e-4, atol=1e-4)


def test_soft_update_real_copy():
    net = MlpLSTMCritic(4, 2, 5, 32)
    target = copy.deepcopy(net)
    with torch.no_grad():
        for p in net.parameters():
            p.add_(1.0)
    soft_update(net, target, tau=0.5)
    for p, tp in zip(net.parameters(), target.parameters()):
        assert not torch.allclose(p, tp)  # NOT aliased (the reference bug)
        torch.testing.assert_close(tp, p - 0.5, rtol=1e-5, atol=1e-5)


def test_kl_divergence_vs_torch():
    torch.manual_seed(3)
    lp = torch.randn(B, T, 3)
    lq = torch.randn(B, T, 3)
    got = kl_divergence(lp, lq).squeeze(-1)
    want = torch.distributions.kl_divergence(
        torch.distributions.Categorical(logits=lp),
        torch.distributions.Categorical(logits=lq),
    )
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)
    assert (kl_divergence(lp, lp).abs() < 1e-6).all()
