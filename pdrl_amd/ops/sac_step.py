"""FusedSacStep: the SAC (discrete) training iteration as a fixed HIP kernel
DAG — the discrete half of K11, hipGraph-capturable.

Follows the EAGER reference ordering exactly
(agents/learner_module/sac/learning.py):
  1.  actor fwd, twin-critic fwds
  2.  sac_actor_loss      — analytic dlogits + dlog_alpha + stats
  3.  actor BPTT + MFMA wgrad → flat actor grads (+ norm) → fused Adam
  4.  alpha fused Adam (no clip)
  5.  actor fwd AGAIN (post-update policy, as the reference does)
  6.  target-critic fwds
  7.  sac_critic_loss     — soft-Q TD target, twin huber, analytic dq1/dq2
  8.  q1/q2 BPTT + wgrads → flat critic grads (+ norm) → fused Adam
  9.  Polyak soft_update (cached device pointer tables)

The twin-critic outputs from step 1 are reused for the value loss — the
reference recomputes them after the actor update, but the critic parameters
are unchanged in between, so the values are identical.
"""
from __future__ import annotations

import torch

from . import ext
from .fused_step import GraphableStep

_SAC_STATS = ["loss-actor", "loss-alpha", "alpha", "entropy", "loss-value"]


class FusedSacStep(GraphableStep):
    def __init__(self, updater, use_graph: bool = True):
        self.u = updater
        self.params = updater.params
        self.grad_reducer = updater.grad_reducer
        self.cores = {
            "actor": updater.actor.core,
            "q1": updater.critic.q1.core,
            "q2": updater.critic.q2.core,
            "t1": updater.target_critic.q1.core,
            "t2": updater.target_critic.q2.core,
        }
        dev = self.cores["actor"].body_w.device
        self.stats_buf = torch.zeros(8, dtype=torch.float32, device=dev)
        self.stat_names = _SAC_STATS
        self.use_graph = use_graph and self.grad_reducer is None
        # actor+alpha Adam updates happen back-to-back at the same DAG
        # point → one multi-group launch (shared device step clock)
        self.adam_aa = None
        if self.grad_reducer is None and getattr(
                updater.actor_optimizer, "shared_clock", False):
            from .optim import AdamMultiGroup

            self.adam_aa = AdamMultiGroup(
                [updater.actor_optimizer, updater.alpha_optimizer])

    def fits(self, batch) -> bool:
        return True  # the loss kernels grid-stride; no LDS shape limit

    # ------------------------------------------------------------------ #
    def _fwd(self, core, x, hx0, cx0):
        e = ext()
        mo, hS, cS, stash = e.seq_lstm_forward(
            x, hx0, cx0, core.body_w, core.body_b, core.w_ih, core.w_hh,
            core.b_g, core.heads_w, core.heads_b,
        )
        return mo, stash

    def _bwd_wgrad(self, core, gouts, stash, x, hx0, cx0, norm):
        e = ext()
        _, _, _, dgates, dxb = e.seq_lstm_backward_core(
            gouts, None, None, stash, x, cx0, core.body_w, core.w_ih,
            core.w_hh, core.heads_w,
        )
        gs = [core.body_w.grad, core.body_b.grad, core.w_ih.grad,
              core.w_hh.grad, core.b_g.grad, core.heads_w.grad,
              core.heads_b.grad]
        assert all(g is not None for g in gs)
        # kernel order: dw_ih, dw_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b
        e.seq_lstm_wgrad_out(x, hx0, stash, dgates, dxb, gouts,
                             gs[2], gs[3], gs[0], gs[1], gs[4], gs[5], gs[6],
                             norm)

    def _opt(self, optimizer):
        """Apply a fused optimizer whose ||grad||² was accumulated by the
        wgrad kernels (single-rank) or must be recomputed (multi-rank)."""
        if self.grad_reducer is not None:
            self.grad_reducer.all_reduce([optimizer.flat_grad])
            optimizer.step()
        else:
            optimizer._update()

    def _body(self, batch):
        from pdrl_amd.agents.learner_module.compute_loss import soft_update

        u, p = self.u, self.params
        e = ext()
        x = batch["obs"]
        B, S, _ = x.shape
        hx0 = batch["hx"][:, 0]
        cx0 = batch["cx"][:, 0]
        act = batch["act"].reshape(-1)
        rew = batch["rew"].reshape(B, S)
        fir = batch["is_fir"].reshape(B, S)
        log_alpha = u.log_alpha.data.view(1)
        single = self.grad_reducer is None

        # 1. actor + twin critic forwards
        moA1, stA = self._fwd(self.cores["actor"], x, hx0, cx0)
        mq1, st1 = self._fwd(self.cores["q1"], x, hx0, cx0)
        mq2, st2 = self._fwd(self.cores["q2"], x, hx0, cx0)

        # 2. actor + temperature losses (analytic grads)
        gA = torch.empty_like(moA1)
        e.sac_actor_loss(
            moA1, mq1, mq2, log_alpha, gA, u.log_alpha.grad.view(1),
            self.stats_buf[:4], u.actor_optimizer.norm_sq if single else None,
            None, u.target_entropy,
        )
        # 3-4. actor + alpha updates
        self._bwd_wgrad(self.cores["actor"], gA, stA, x, hx0, cx0,
                        u.actor_optimizer.norm_sq if single else None)
        if self.adam_aa is not None:
            self.adam_aa.update()  # actor + alpha in ONE launch
        else:
            self._opt(u.actor_optimizer)
            if self.grad_reducer is not None:
                self.grad_reducer.all_reduce([u.alpha_optimizer.flat_grad])
            u.alpha_optimizer._update()  # no clip: norm unused

        # 5-6. post-update actor + target critics
        moA2, _ = self._fwd(self.cores["actor"], x, hx0, cx0)
        mt1, _ = self._fwd(self.cores["t1"], x, hx0, cx0)
        mt2, _ = self._fwd(self.cores["t2"], x, hx0, cx0)

        # 7. critic losses
        gq1 = torch.empty_like(mq1)
        gq2 = torch.empty_like(mq2)
        e.sac_critic_loss(
            moA2, mq1, mq2, mt1, mt2, act, rew, fir, log_alpha, gq1, gq2,
            self.stats_buf[4:5], u.critic_optimizer.norm_sq if single else None,
            p.gamma, p.reward_scale,
        )
        # 8. critic updates (both cores accumulate into one flat space)
        self._bwd_wgrad(self.cores["q1"], gq1, st1, x, hx0, cx0,
                        u.critic_optimizer.norm_sq if single else None)
        self._bwd_wgrad(self.cores["q2"], gq2, st2, x, hx0, cx0,
                        u.critic_optimizer.norm_sq if single else None)
        self._opt(u.critic_optimizer)

        # 9. Polyak target update
        soft_update(u.critic, u.target_critic, u.TAU)

    def _full(self, batch):
        for _ in range(self.params.K_epoch):
            self._body(batch)
