"""V-MPO: on-policy maximum-a-posteriori policy optimization with
temperature (eta) and KL (alpha) Lagrange duals.

Capability parity with the reference's agents/learner_module/v_mpo/learning.py
(GAE + top-50% advantage selection: 60-64; psi-weighted policy loss: 66-74;
temperature dual on eta: 82-85; KL-constraint dual on alpha with sampled
coefficient: 87-92; trainable log_eta/log_alpha in the optimizer —
learner.py:320-348).
"""
from __future__ import annotations

import numpy as np
import torch
import torch.nn.functional as F

from .compute_loss import compute_gae, kl_divergence
from .common import BaseUpdater, batch_initial_state


class VMPOUpdater(BaseUpdater):
    name = "V-MPO"

    def __init__(self, model, params, device, grad_reducer=None):
        super().__init__(params, device, grad_reducer)
        self.model = model.to(device)
        init = float(np.log(params.v_mpo_lagrange_multiplier_init))
        self.log_eta = torch.nn.Parameter(torch.tensor(init, device=self.device))
        self.log_alpha = torch.nn.Parameter(torch.tensor(init, device=self.device))
        self.optimizer = self.make_optimizer(
            "rmsprop",
            list(self.model.parameters()) + [self.log_eta, self.log_alpha],
            lr=params.lr,
            eps=1e-5,
        )
        self.fused_step = self.make_fused_step(
            "V-MPO", self.model, self.optimizer,
            duals=(self.log_eta, self.log_alpha),
        )
        self._rng = np.random.default_rng()

    def trainable_modules(self):
        return {"model": self.model}

    def optimizers(self):
        return {"optimizer": self.optimizer}

    def extra_params(self):
        return {"log_eta": self.log_eta, "log_alpha": self.log_alpha}

    def get_coef_alpha(self):
        """KL-bound coefficient sampled uniformly in [below, upper]
        (reference: learner.py:340-348). Sampled ON DEVICE so a hipGraph
        replay of the step keeps re-sampling (host RNG would freeze)."""
        p = self.params
        u = torch.rand((), device=self.device)
        return p.coef_alpha_below + (p.coef_alpha_upper - p.coef_alpha_below) * u

    def compute_losses(self, batch: dict[str, torch.Tensor]):
        p = self.params
        obs, act = batch["obs"], batch["act"]
        rew = batch["rew"] * p.reward_scale
        behav_logits, is_fir = batch["logits"], batch["is_fir"]
        hx0, cx0 = batch_initial_state(batch)

        logits, log_probs, _, value = self.model.actor(obs, (hx0, cx0), act)

        with torch.no_grad():
            mask = 1.0 - is_fir[:, 1:]
            td_target = rew[:, :-1] + p.gamma * mask * value[:, 1:]
            delta = td_target - value[:, :-1]
            adv = compute_gae(delta, p.gamma, p.lmbda, dones=is_fir[:, 1:])

        eta = self.log_eta.exp()
        alpha = self.log_alpha.exp()

        # top-half advantage selection over the flattened batch
        flat_adv = adv.reshape(-1)
        flat_logp = log_probs[:, :-1].reshape(-1)
        k = max(1, flat_adv.numel() // 2)
        top_adv, top_idx = torch.topk(flat_adv, k)
        top_logp = flat_logp[top_idx]

        # psi-weighted MAP policy loss
        psi = torch.softmax(top_adv / eta.detach(), dim=0)
        policy_loss = -(psi * top_logp).sum()

        # temperature dual: eta* minimizes eta*eps + eta*log E[exp(adv/eta)]
        eps_eta = p.coef_eta
        eta_loss = eta * eps_eta + eta * (
            torch.logsumexp(top_adv / eta, dim=0) - float(np.log(k))
        )

        # KL-constraint dual on alpha between behaviour and target policies
        kl = kl_divergence(behav_logits[:, :-1], logits[:, :-1]).mean()
        eps_alpha = self.get_coef_alpha()
        alpha_loss = alpha * (eps_alpha - kl.detach()) + alpha.detach() * kl

        value_loss = F.smooth_l1_loss(value[:, :-1], td_target)

        loss = (
            p.policy_loss_coef * policy_loss
            + p.value_loss_coef * value_loss
            + eta_loss
            + alpha_loss
        )
        stats = {
            "loss-total": loss.detach(),
            "loss-policy": policy_loss.detach(),
            "loss-value": value_loss.detach(),
            "eta": eta.detach(),
            "alpha": alpha.detach(),
            "kl": kl.detach(),
        }
        return loss, stats

    def step(self, batch: dict[str, torch.Tensor]) -> dict:
        if self.fused_step is not None and self.fused_step.fits(batch):
            stats = self.fused_step.run(batch)
            self.update_count += 1
            return stats
        stats = {}
        for _ in range(self.params.K_epoch):
            loss, stats = self.compute_losses(batch)
            self.optimizer.zero_grad(set_to_none=False)
            loss.backward()
            self.apply_step(
                self.optimizer,
                list(self.model.parameters()) + [self.log_eta, self.log_alpha],
            )
        self.update_count += 1
        return stats
