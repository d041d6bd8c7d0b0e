"""PPO: clipped-surrogate policy optimization over LSTM sequence batches.

Capability parity with the reference's agents/learner_module/ppo/learning.py
(TD target + GAE under no_grad: 48-57; clipped ratio surrogate + smooth-L1
value loss + entropy bonus: 59-81; clip_grad_norm + RMSprop: 90-106).
Redesigned as a pure ``step(batch)`` updater so the same math runs under CPU
tests, the eager GPU oracle, and the hipGraph-captured fused path.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from .compute_loss import compute_gae
from .common import BaseUpdater, batch_initial_state


class PPOUpdater(BaseUpdater):
    name = "PPO"

    def __init__(self, model, params, device, grad_reducer=None):
        super().__init__(params, device, grad_reducer)
        self.model = model.to(device)
        self.optimizer = self.make_optimizer(
            "rmsprop", self.model.parameters(), lr=params.lr, eps=1e-5
        )
        self.fused_step = self.make_fused_step("PPO", self.model, self.optimizer)

    def trainable_modules(self):
        return {"model": self.model}

    def optimizers(self):
        return {"optimizer": self.optimizer}

    def compute_losses(self, batch: dict[str, torch.Tensor]):
        p = self.params
        obs, act = batch["obs"], batch["act"]
        rew = batch["rew"] * p.reward_scale
        behav_log_prob, is_fir = batch["log_prob"], batch["is_fir"]
        hx0, cx0 = batch_initial_state(batch)

        logits, log_probs, entropy, value = self.model.actor(obs, (hx0, cx0), act)

        with torch.no_grad():
            mask = 1.0 - is_fir[:, 1:]
            td_target = rew[:, :-1] + p.gamma * mask * value[:, 1:]
            delta = td_target - value[:, :-1]
            gae = compute_gae(delta, p.gamma, p.lmbda, dones=is_fir[:, 1:])

        ratio = torch.exp(log_probs[:, :-1] - behav_log_prob[:, :-1])
        surr1 = ratio * gae
        surr2 = torch.clamp(ratio, 1.0 - p.eps_clip, 1.0 + p.eps_clip) * gae
        policy_loss = -torch.min(surr1, surr2).mean()
        value_loss = F.smooth_l1_loss(value[:, :-1], td_target)
        entropy_mean = entropy[:, :-1].mean()

        # logit L2 keeps the policy out of exact one-hot saturation (an fp32
        # one-hot has ZERO policy/entropy gradients — an absorbing collapse
        # state observed at high update rates)
        logit_reg = float(getattr(p, "logit_reg", 0.0))
        reg_loss = logits[:, :-1].pow(2).mean() if logit_reg > 0 else 0.0
        loss = (
            p.policy_loss_coef * policy_loss
            + p.value_loss_coef * value_loss
            - p.entropy_coef * entropy_mean
            + logit_reg * reg_loss
        )
        stats = {
            "loss-total": loss.detach(),
            "loss-policy": policy_loss.detach(),
            "loss-value": value_loss.detach(),
            "entropy": entropy_mean.detach(),
            "ratio-avg": ratio.detach().mean(),
            "ratio-min": ratio.detach().min(),
            "ratio-max": ratio.detach().max(),
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
            self.apply_step(self.optimizer, self.model.parameters())
        self.update_count += 1
        return stats
