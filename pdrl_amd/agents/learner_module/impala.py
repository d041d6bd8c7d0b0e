"""IMPALA: V-trace off-policy actor-critic.

Capability parity with the reference's agents/learner_module/impala/learning.py
(V-trace correction: 48-57; policy-gradient loss -(log π · adv) + value loss
to vs targets + entropy bonus: 59-69). This is the flagship benchmark
algorithm (BASELINE.json configs[1]).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from .compute_loss import compute_v_trace
from .common import BaseUpdater, batch_initial_state


class ImpalaUpdater(BaseUpdater):
    name = "IMPALA"

    def __init__(self, model, params, device, grad_reducer=None):
        super().__init__(params, device, grad_reducer)
        self.model = model.to(device)
        self.optimizer = self.make_optimizer(
            "rmsprop", self.model.parameters(), lr=params.lr, eps=1e-5
        )
        self.fused_step = self.make_fused_step("IMPALA", self.model, self.optimizer)

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

        rhos, advantages, values_target = compute_v_trace(
            behav_log_prob, log_probs, is_fir, rew, value, p.gamma
        )

        policy_loss = -(log_probs[:, :-1] * advantages).mean()
        value_loss = F.smooth_l1_loss(value[:, :-1], values_target)
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
            "rho-avg": rhos.mean(),
        }
        return loss, stats

    def step(self, batch: dict[str, torch.Tensor]) -> dict:
        if self.fused_step is not None:
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
