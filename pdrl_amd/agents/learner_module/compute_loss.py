"""Shared RL loss math: GAE, V-trace, Polyak update, categorical KL.

Capability parity with the reference's agents/learner_module/compute_loss.py
(compute_gae:7-19, compute_v_trace:22-66, soft_update:69-71,
kldivergence:74-77). The eager implementations here are the numerics oracle;
on GPU the same entry points dispatch to fused CDNA4 HIP scan kernels
(K6/K7/K12 in SURVEY.md §2.4) when the extension is loaded.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F


_soft_update_tables: dict = {}


def _use_hip(*tensors) -> bool:
    if not all(t.is_cuda for t in tensors if isinstance(t, torch.Tensor)):
        return False
    from pdrl_amd import ops

    return ops.available()


def compute_gae(
    deltas: torch.Tensor, gamma: float, lmbda: float, dones: torch.Tensor | None = None
) -> torch.Tensor:
    """Generalized advantage estimation reverse scan.

    deltas: (B, T, 1) TD residuals; dones (B, T, 1) optional mask that zeroes
    the recursion across episode boundaries. Returns advantages (B, T, 1):
        adv_t = delta_t + gamma*lmbda*(1 - done_t) * adv_{t+1}
    """
    if _use_hip(deltas):
        from pdrl_amd import ops

        d = dones.contiguous() if dones is not None else torch.zeros_like(deltas)
        return ops.ext().gae(deltas.contiguous(), float(gamma), float(lmbda), d)
    B, T, _ = deltas.shape
    adv = torch.zeros_like(deltas)
    running = torch.zeros(B, 1, dtype=deltas.dtype, device=deltas.device)
    for t in reversed(range(T)):
        mask = 1.0 - (dones[:, t] if dones is not None else 0.0)
        running = deltas[:, t] + gamma * lmbda * mask * running
        adv[:, t] = running
    return adv


def compute_v_trace(
    behav_log_probs: torch.Tensor,
    target_log_probs: torch.Tensor,
    is_fir: torch.Tensor,
    rewards: torch.Tensor,
    values: torch.Tensor,
    gamma: float,
    rho_bar: float = 0.8,
    rho_min: float = 0.1,
    c_bar: float = 1.0,
):
    """V-trace off-policy correction (IMPALA; Espeholt et al. 2018), with the
    reference's clamp band rho ∈ [0.1, 0.8], c̄=1.0
    (reference: compute_loss.py:22-66).

    All inputs (B, S, 1); uses timesteps 0..S-2 for targets. ``is_fir`` marks
    the first step of an episode (resets bootstrapping via mask = 1-is_fir of
    the NEXT step). Returns (rhos, advantages, values_target) each (B, S-1, 1).
    """
    if _use_hip(behav_log_probs, target_log_probs, rewards, values):
        from pdrl_amd import ops

        return ops.ext().vtrace(
            behav_log_probs.contiguous(), target_log_probs.contiguous(),
            is_fir.contiguous(), rewards.contiguous(), values.contiguous(),
            float(gamma), float(rho_bar), float(rho_min), float(c_bar),
        )
    log_rhos = (target_log_probs - behav_log_probs)[:, :-1]
    rhos = torch.clamp(log_rhos.exp(), rho_min, rho_bar)
    cs = torch.clamp(log_rhos.exp(), max=c_bar)
    mask = 1.0 - is_fir[:, 1:]  # no bootstrap across episode starts

    v_cur = values[:, :-1]
    v_next = values[:, 1:]
    r = rewards[:, :-1]
    deltas = rhos * (r + gamma * mask * v_next - v_cur)

    B, T, _ = deltas.shape
    acc = torch.zeros(B, 1, dtype=values.dtype, device=values.device)
    vs_minus_v = torch.zeros_like(deltas)
    for t in reversed(range(T)):
        acc = deltas[:, t] + gamma * mask[:, t] * cs[:, t] * acc
        vs_minus_v[:, t] = acc
    vs = v_cur + vs_minus_v
    vs_next = torch.cat([vs[:, 1:], v_next[:, -1:]], dim=1)
    advantages = rhos * (r + gamma * mask * vs_next - v_cur)
    return rhos.detach(), advantages.detach(), vs.detach()


@torch.no_grad()
def soft_update(net: torch.nn.Module, target_net: torch.nn.Module, tau: float = 0.005):
    """Polyak target update θ' ← (1-τ)θ' + τθ (reference: compute_loss.py:69-71).
    On GPU with the extension loaded this is one fused multi-tensor axpby
    kernel (K12) instead of a per-parameter op loop."""
    params = list(net.parameters())
    tparams = list(target_net.parameters())
    assert len(params) == len(tparams)
    if params and _use_hip(*params, *tparams):
        from pdrl_amd import ops

        key = (id(net), id(target_net))
        tab = _soft_update_tables.get(key)
        # revalidate BOTH sides: param storages move when an optimizer (or
        # the fused SAC step's Polyak flat buffer) re-homes them
        ptrs = [p.data.data_ptr() for p in params] + \
               [t.data.data_ptr() for t in tparams]
        if tab is None or tab[3] != ptrs:
            dev = params[0].device
            sp = torch.tensor([p.data.data_ptr() for p in params],
                              dtype=torch.int64).to(dev)
            dp = torch.tensor([t.data.data_ptr() for t in tparams],
                              dtype=torch.int64).to(dev)
            ne = torch.tensor([p.numel() for p in params],
                              dtype=torch.int64).to(dev)
            mx = max(p.numel() for p in params)
            tab = (sp, dp, ne, ptrs, mx)
            _soft_update_tables[key] = tab
        ops.ext().soft_update_cached(tab[0], tab[1], tab[2], len(params),
                                     tab[4], float(tau))
        return
    for p, tp in zip(params, tparams):
        tp.data.mul_(1.0 - tau).add_(p.data, alpha=tau)


def kl_divergence(logits_p: torch.Tensor, logits_q: torch.Tensor) -> torch.Tensor:
    """KL(P || Q) between categorical distributions given by logits
    (reference: compute_loss.py:74-77). Returns (..., 1)."""
    logp = F.log_softmax(logits_p, dim=-1)
    logq = F.log_softmax(logits_q, dim=-1)
    return (logp.exp() * (logp - logq)).sum(-1, keepdim=True)
