"""Analytic-gradient derivation for the round-2 SAC-Continuous fused loss
kernels (docs/ROUND1_NOTES.md), verified against autograd on CPU — the same
derive-then-kernelize workflow used for the V-MPO / PPO-C mega kernels.

Actor loss (reparameterized, per element; eps fixed by the reparam trick):
    z = mu + std*eps,  a = tanh(z),  std = exp(clamp(ls, -20, 2))
    logpi = sum_j [ -0.5*eps_j^2 - ls_j - 0.5*log(2*pi) - log(1 - a_j^2 + 1e-7) ]
    L_actor = mean( alpha * logpi - min(Q1(s,a), Q2(s,a)) )
    L_alpha = -mean( log_alpha * (logpi.detach() + target_entropy) )

With g = dmin(Q1,Q2)/da_j (the cross-network input gradient the critic
core's backward dx provides) and t = 2*a*(1-a^2)/(1-a^2+1e-7):
    dL/dmu_j = (1/N) * ( alpha * t  -  g * (1-a^2) )
    dL/dls_j = (1/N) * ( alpha * (-1 + t*std*eps) - g*(1-a^2)*std*eps ) * m
        where m = 1 inside the clamp band, 0 outside
    dL/dlog_alpha = -mean(logpi + target_entropy)

Critic loss (post-actor-update policy sample a', logpi'):
    v_next = min(tQ1, tQ2)[:,1:] - alpha * logpi'[:,1:]
    y = r_scaled[:, :-1] + gamma * (1 - is_fir[:,1:]) * v_next
    L_value = huber(q1[:,:-1], y) + huber(q2[:,:-1], y)
    dL/dq_{b,s} = huber'(q - y) / N'   (s < S-1; 0 at the last step)
"""
import math

import pytest
import torch

LOG_STD_MIN, LOG_STD_MAX = -20.0, 2.0
EPS_A = 1e-7


def _actor_forward(mu, ls_raw, eps):
    ls = ls_raw.clamp(LOG_STD_MIN, LOG_STD_MAX)
    std = ls.exp()
    z = mu + std * eps
    a = torch.tanh(z)
    logpi = (
        -0.5 * eps.pow(2) - ls - 0.5 * math.log(2 * math.pi)
        - torch.log(1.0 - a.pow(2) + EPS_A)
    ).sum(-1, keepdim=True)
    return a, logpi


def _analytic_actor_grads(mu, ls_raw, eps, g, alpha, N):
    """The formulas the round-2 HIP kernel will implement."""
    ls = ls_raw.clamp(LOG_STD_MIN, LOG_STD_MAX)
    std = ls.exp()
    z = mu + std * eps
    a = torch.tanh(z)
    one_m_a2 = 1.0 - a.pow(2)
    t = 2.0 * a * one_m_a2 / (one_m_a2 + EPS_A)
    dmu = (alpha * t - g * one_m_a2) / N
    m = ((ls_raw > LOG_STD_MIN) & (ls_raw < LOG_STD_MAX)).float()
    dls = (alpha * (-1.0 + t * std * eps) - g * one_m_a2 * std * eps) * m / N
    return dmu, dls


class _TinyCritic(torch.nn.Module):
    """Stand-in for the critic stack: any smooth Q(s, a)."""

    def __init__(self, f, A, seed):
        super().__init__()
        torch.manual_seed(seed)
        self.w1 = torch.nn.Linear(f + A, 16)
        self.w2 = torch.nn.Linear(16, 1)

    def forward(self, obs, act):
        return self.w2(torch.tanh(self.w1(torch.cat([obs, act], -1))))


@pytest.mark.parametrize("shape", [(4, 5, 3, 2), (2, 7, 6, 1)])
def test_sacc_actor_analytic_grads_vs_autograd(shape):
    B, S, F, A = shape
    torch.manual_seed(0)
    obs = torch.randn(B, S, F, dtype=torch.float64)
    mu = torch.randn(B, S, A, dtype=torch.float64, requires_grad=True)
    # include values outside the clamp band to exercise the mask
    ls_raw = (torch.randn(B, S, A, dtype=torch.float64) * 2.0).requires_grad_(True)
    ls_raw.data[0, 0] = 3.0  # clamped high
    eps = torch.randn(B, S, A, dtype=torch.float64)
    log_alpha = torch.tensor(-0.7, dtype=torch.float64, requires_grad=True)
    q1 = _TinyCritic(F, A, 1).double()
    q2 = _TinyCritic(F, A, 2).double()
    target_entropy = -float(A)
    N = B * S

    a, logpi = _actor_forward(mu, ls_raw, eps)
    qmin = torch.min(q1(obs, a), q2(obs, a))
    actor_loss = (log_alpha.exp().detach() * logpi - qmin).mean()
    actor_loss.backward()

    # the cross-network piece the kernel receives from the critic-core
    # backward's dx (here: autograd of qmin wrt a alone)
    a2, _ = _actor_forward(mu.detach(), ls_raw.detach(), eps)
    a2.requires_grad_(True)
    torch.min(q1(obs, a2), q2(obs, a2)).sum().backward()
    g = a2.grad

    dmu, dls = _analytic_actor_grads(
        mu.detach(), ls_raw.detach(), eps, g, log_alpha.exp().item(), N)
    torch.testing.assert_close(dmu, mu.grad, rtol=1e-9, atol=1e-10)
    torch.testing.assert_close(dls, ls_raw.grad, rtol=1e-9, atol=1e-10)

    # temperature
    alpha_loss = -(log_alpha * (logpi.detach() + target_entropy)).mean()
    log_alpha.grad = None
    alpha_loss.backward()
    dlog_alpha = -(logpi.detach() + target_entropy).mean()
    torch.testing.assert_close(dlog_alpha, log_alpha.grad, rtol=1e-9, atol=1e-12)


def test_sacc_critic_analytic_grads_vs_autograd():
    B, S = 4, 6
    torch.manual_seed(3)
    q1 = torch.randn(B, S, 1, dtype=torch.float64, requires_grad=True)
    q2 = torch.randn(B, S, 1, dtype=torch.float64, requires_grad=True)
    tq = torch.randn(B, S, 1, dtype=torch.float64) * 3
    logpi_next = torch.randn(B, S, 1, dtype=torch.float64)
    rew = torch.randn(B, S, 1, dtype=torch.float64)
    fir = (torch.rand(B, S, 1) < 0.2).double()
    alpha, gamma, scale = 0.21, 0.997, 1.0

    v_next = tq[:, 1:] - alpha * logpi_next[:, 1:]
    y = scale * rew[:, :-1] + gamma * (1.0 - fir[:, 1:]) * v_next
    loss = torch.nn.functional.smooth_l1_loss(q1[:, :-1], y) + \
        torch.nn.functional.smooth_l1_loss(q2[:, :-1], y)
    loss.backward()

    Np = B * (S - 1)
    for q, qgrad in ((q1, q1.grad), (q2, q2.grad)):
        err = (q.detach()[:, :-1] - y)
        dq = torch.where(err.abs() < 1.0, err, err.sign()) / Np
        full = torch.zeros_like(q)
        full[:, :-1] = dq
        torch.testing.assert_close(full, qgrad, rtol=1e-9, atol=1e-12)
