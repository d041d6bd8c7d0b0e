"""SAC (continuous actions) with reparameterized tanh-Gaussian actor.

Capability parity with the reference's
agents/learner_module/sac_continuous/learning.py (reparameterized actor loss
alpha*log pi - min Q: 44-55; critic on (obs, act): 94-101; alpha auto-tune;
Polyak soft update). Target critic is a real deep copy (the reference's
aliasing bug is fixed — see sac.py docstring).
"""
from __future__ import annotations

import copy

import numpy as np
import torch
import torch.nn.functional as F

from .compute_loss import soft_update
from .common import BaseUpdater, batch_initial_state


class SACContinuousUpdater(BaseUpdater):
    name = "SAC-Continuous"
    TAU = 0.005

    def __init__(self, model, params, device, grad_reducer=None):
        super().__init__(params, device, grad_reducer)
        self.model = model.to(device)
        self.actor = self.model.actor
        self.critic = self.model.critic
        self.target_critic = copy.deepcopy(self.critic).to(device)
        for p in self.target_critic.parameters():
            p.requires_grad_(False)

        n_actions = self.actor.n_outputs
        # default -dim(A) (the SAC heuristic); overridable — on sparse
        # exploration tasks a lower target lets alpha decay slower.
        # `is not None` (not `or`): an explicit 0.0 is a valid target.
        te = getattr(params, "target_entropy", None)
        self.target_entropy = float(te) if te is not None else -float(n_actions)
        self.log_alpha = torch.nn.Parameter(
            torch.tensor(float(np.log(params.alpha)), device=self.device)
        )

        # one shared device Adam step clock across the three optimizers
        # (they always step together); the actor optimizer owns the clock
        # and steps first in both the eager and the fused ordering
        clock = torch.zeros(3, dtype=torch.float32, device=self.device) \
            if self.device.type == "cuda" else None
        self.actor_optimizer = self.make_optimizer(
            "adam", self.actor.parameters(), lr=params.lr, clock=clock,
            clock_owner=True)
        self.critic_optimizer = self.make_optimizer(
            "adam", self.critic.parameters(), lr=params.lr, clock=clock)
        self.alpha_optimizer = self.make_optimizer(
            "adam", [self.log_alpha], lr=params.lr, clip=False, clock=clock)
        self.fused_step = self._make_fused_step()

    def _make_fused_step(self):
        """Whole-step fused HIP DAG (ops/sacc_step.py; kernel math verified
        vs autograd in tests/test_sacc_analytic.py)."""
        import os

        from pdrl_amd import ops

        if not (self.device.type == "cuda" and ops.available()):
            return None
        if not all(getattr(o, "is_fused", False) for o in
                   (self.actor_optimizer, self.critic_optimizer,
                    self.alpha_optimizer)):
            return None
        if self.actor.core.head_names != ["mu", "log_std"]:
            return None
        if not bool(int(os.environ.get("PDRL_SACC_FUSED", "1"))):
            return None
        from pdrl_amd.ops.sacc_step import FusedSacContinuousStep

        use_graph = bool(int(os.environ.get("PDRL_USE_GRAPH", "1")))
        return FusedSacContinuousStep(self, use_graph=use_graph)

    def trainable_modules(self):
        return {"model": self.model, "target_critic": self.target_critic}

    def extra_params(self):
        return {"log_alpha": self.log_alpha}

    def optimizers(self):
        return {
            "actor_optimizer": self.actor_optimizer,
            "critic_optimizer": self.critic_optimizer,
            "alpha_optimizer": self.alpha_optimizer,
        }

    @property
    def alpha(self):
        return self.log_alpha.exp()

    def step(self, batch: dict[str, torch.Tensor]) -> dict:
        if self.fused_step is not None and self.fused_step.fits(batch):
            stats = self.fused_step.run(batch)
            self.update_count += 1
            return stats
        p = self.params
        obs, act = batch["obs"], batch["act"]
        rew = batch["rew"] * p.reward_scale
        is_fir = batch["is_fir"]
        hx0, cx0 = batch_initial_state(batch)
        stats = {}

        for _ in range(p.K_epoch):
            # -- actor (reparameterized) --------------------------------- #
            new_act, log_prob = self.actor(obs, (hx0, cx0))
            q1_pi, q2_pi = self.critic(obs, new_act, (hx0, cx0))
            actor_loss = (self.alpha.detach() * log_prob - torch.min(q1_pi, q2_pi)).mean()
            self.actor_optimizer.zero_grad(set_to_none=False)
            actor_loss.backward()
            self.apply_step(self.actor_optimizer, self.actor.parameters())

            # -- temperature --------------------------------------------- #
            alpha_loss = -(
                self.log_alpha * (log_prob.detach() + self.target_entropy)
            ).mean()
            self.alpha_optimizer.zero_grad(set_to_none=False)
            alpha_loss.backward()
            if getattr(self.alpha_optimizer, "is_fused", False):
                if self.grad_reducer is not None:
                    self.grad_reducer.all_reduce([self.alpha_optimizer.flat_grad])
                self.alpha_optimizer.step()
            else:
                if self.grad_reducer is not None:
                    self.grad_reducer.all_reduce([self.log_alpha.grad])
                self.alpha_optimizer.step()

            # -- critics -------------------------------------------------- #
            with torch.no_grad():
                next_act, next_log_prob = self.actor(obs, (hx0, cx0))
                tq1, tq2 = self.target_critic(obs, next_act, (hx0, cx0))
                v_next = torch.min(tq1, tq2)[:, 1:] - self.alpha * next_log_prob[:, 1:]
                mask = 1.0 - is_fir[:, 1:]
                target_q = rew[:, :-1] + p.gamma * mask * v_next

            q1, q2 = self.critic(obs, act, (hx0, cx0))
            value_loss = F.smooth_l1_loss(q1[:, :-1], target_q) + F.smooth_l1_loss(
                q2[:, :-1], target_q
            )
            self.critic_optimizer.zero_grad(set_to_none=False)
            value_loss.backward()
            self.apply_step(self.critic_optimizer, self.critic.parameters())

            soft_update(self.critic, self.target_critic, self.TAU)

            stats = {
                "loss-actor": actor_loss.detach(),
                "loss-value": value_loss.detach(),
                "loss-alpha": alpha_loss.detach(),
                "alpha": self.alpha.detach(),
            }
        self.update_count += 1
        return stats
