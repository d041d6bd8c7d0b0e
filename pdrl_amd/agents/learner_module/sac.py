"""SAC (discrete actions, expectation form) with auto-tuned entropy
temperature and a genuinely deep-copied target critic.

Capability parity with the reference's agents/learner_module/sac/learning.py
(expectation-form actor loss: 36-62; alpha auto-tune vs target entropy:
64-74; soft-Q TD target from the target critic: 76-103; twin-Q smooth-L1
value loss gathered on the taken action: 105-120; Polyak soft update: 143)
and learner.py:351-400 (three optimizers + log_alpha + latching replay-ready
flag). The reference's target-critic aliasing bug (learner.py:357 — ``.to()``
returns the same module, making soft_update a no-op) is fixed here via
copy.deepcopy.
"""
from __future__ import annotations

import copy

import numpy as np
import torch
import torch.nn.functional as F

from .compute_loss import soft_update
from .common import BaseUpdater, batch_initial_state


class SACUpdater(BaseUpdater):
    name = "SAC"
    TAU = 0.005

    def __init__(self, model, params, device, grad_reducer=None):
        super().__init__(params, device, grad_reducer)
        self.model = model.to(device)
        self.actor = self.model.actor
        self.critic = self.model.critic
        self.target_critic = copy.deepcopy(self.critic).to(device)
        for p in self.target_critic.parameters():
            p.requires_grad_(False)

        n_actions = self.critic.n_outputs
        # maximum-entropy target: 98% of uniform-policy entropy (the
        # reference's choice, sac/learning.py) — overridable: a target this
        # close to uniform makes alpha explode on easy tasks, pinning the
        # policy near-random (measured: CartPole capped at ~110 reward with
        # alpha ~9; target 0.35 nats solves it)
        te = getattr(params, "target_entropy", None)
        self.target_entropy = float(te) if te is not None else \
            0.98 * float(-np.log(1.0 / n_actions))
        self.log_alpha = torch.nn.Parameter(
            torch.tensor(float(np.log(params.alpha)), device=self.device)
        )

        # the three optimizers always step together once per iteration →
        # they share ONE device step clock (t, bc1, bc2). The actor
        # optimizer owns the clock and steps FIRST in every path (eager
        # order actor→alpha→critic; fused DAG ticks at the actor update),
        # so alpha/critic always see a freshly advanced bias correction.
        clock = torch.zeros(3, dtype=torch.float32, device=self.device) \
            if self.device.type == "cuda" else None
        self.actor_optimizer = self.make_optimizer(
            "adam", self.actor.parameters(), lr=params.lr, clock=clock,
            clock_owner=True)
        self.critic_optimizer = self.make_optimizer(
            "adam", self.critic.parameters(), lr=params.lr, clock=clock)
        self.alpha_optimizer = self.make_optimizer(
            "adam", [self.log_alpha], lr=params.lr, clip=False, clock=clock)
        self.fused_step = self._make_sac_fused_step()

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

    def _make_sac_fused_step(self):
        """Whole-step fused HIP DAG for discrete SAC (ops/sac_step.py)."""
        from pdrl_amd import ops

        if not (self.device.type == "cuda" and ops.available()):
            return None
        if not all(getattr(o, "is_fused", False) for o in
                   (self.actor_optimizer, self.critic_optimizer,
                    self.alpha_optimizer)):
            return None
        if getattr(self.actor, "core", None) is None or \
                self.actor.core.head_names != ["logits"]:
            return None
        import os

        from pdrl_amd.ops.sac_step import FusedSacStep

        use_graph = bool(int(os.environ.get("PDRL_USE_GRAPH", "1")))
        return FusedSacStep(self, use_graph=use_graph)

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
            probs, log_probs = self.actor(obs, (hx0, cx0))
            q1, q2 = self.critic(obs, (hx0, cx0))

            # -- actor: E_a~pi[ alpha*log pi - min Q ] ------------------- #
            min_q = torch.min(q1, q2).detach()
            actor_loss = (
                (probs * (self.alpha.detach() * log_probs - min_q)).sum(-1).mean()
            )
            self.actor_optimizer.zero_grad(set_to_none=False)
            actor_loss.backward()
            self.apply_step(self.actor_optimizer, self.actor.parameters())

            # -- temperature auto-tune ----------------------------------- #
            with torch.no_grad():
                pi_entropy = -(probs * log_probs).sum(-1)
            alpha_loss = (self.log_alpha * (pi_entropy - self.target_entropy).detach()).mean()
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

            # -- critics: soft-Q TD target ------------------------------- #
            with torch.no_grad():
                next_probs, next_log_probs = self.actor(obs, (hx0, cx0))
                tq1, tq2 = self.target_critic(obs, (hx0, cx0))
                v_next = (
                    next_probs[:, 1:]
                    * (torch.min(tq1, tq2)[:, 1:] - self.alpha * next_log_probs[:, 1:])
                ).sum(-1, keepdim=True)
                mask = 1.0 - is_fir[:, 1:]
                target_q = rew[:, :-1] + p.gamma * mask * v_next

            q1c, q2c = self.critic(obs, (hx0, cx0))
            a_idx = act[:, :-1].long()
            q1_taken = q1c[:, :-1].gather(-1, a_idx)
            q2_taken = q2c[:, :-1].gather(-1, a_idx)
            value_loss = F.smooth_l1_loss(q1_taken, target_q) + F.smooth_l1_loss(
                q2_taken, target_q
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
                "entropy": pi_entropy.mean(),
            }
        self.update_count += 1
        return stats
