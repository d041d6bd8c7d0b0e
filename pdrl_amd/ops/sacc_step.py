"""FusedSacContinuousStep: the SAC-Continuous training iteration as a fixed
HIP kernel DAG — the continuous half of K11,
hipGraph-capturable.

Follows the EAGER reference ordering exactly
(agents/learner_module/sac_continuous/learning.py, reference
sac_continuous/learning.py:13-151):
  1.  actor fwd → reparameterized tanh-Gaussian sample (sacc_sample:
      in-kernel counter RNG, graph-replay safe) → a_new, log pi
  2.  twin critics on (obs, a_new); d minQ/da via the critic cores'
      backward dx routed back through the action encoder (the
      cross-network gradient)
  3.  sacc_actor_grad — analytic dmu/dlog_std (verified vs autograd,
      tests/test_sacc_analytic.py) + dlog_alpha + stats
  4.  actor BPTT + MFMA wgrad → fused Adam; alpha Adam
  5.  actor fwd AGAIN (post-update sample a', log pi')
  6.  target critics on (obs, a'); behavior critics on (obs, batch act)
  7.  sacc_critic_loss — soft-Q target, twin huber, analytic dq1/dq2
  8.  critic BPTT + wgrads (cores via MFMA kernels; the obs/act encoders
      via library GEMMs into the flat-grad views) → fused Adam
  9.  Polyak soft_update (cached device pointer tables)

The critic stacks are MlpLSTMCriticContinuous: enc = [relu(obs@Wo+bo) |
relu(act@Wa+ba)] feeding the LSTM core (networks/models.py) — the encoder
fwd/bwd are single library GEMMs and stay as torch ops inside the captured
graph.
"""
from __future__ import annotations

import torch

from . import ext
from .fused_step import GraphableStep

_SACC_STATS = ["loss-actor", "loss-alpha", "alpha", "entropy", "loss-value"]


class FusedSacContinuousStep(GraphableStep):
    def __init__(self, updater, use_graph: bool = True):
        self.u = updater
        self.params = updater.params
        self.grad_reducer = updater.grad_reducer
        self.actor_core = updater.actor.core
        self.q = {
            "q1": updater.critic.q1,
            "q2": updater.critic.q2,
            "t1": updater.target_critic.q1,
            "t2": updater.target_critic.q2,
        }
        dev = self.actor_core.body_w.device
        self.stats_buf = torch.zeros(8, dtype=torch.float32, device=dev)
        self.stat_names = _SACC_STATS
        self.rng = torch.randint(1, 1 << 30, (1,), dtype=torch.int32, device=dev)
        self.use_graph = use_graph and self.grad_reducer is None
        # actor+alpha Adam updates batch into one launch (shared clock)
        self.adam_aa = None
        if self.grad_reducer is None and getattr(
                updater.actor_optimizer, "shared_clock", False):
            from .optim import AdamMultiGroup

            self.adam_aa = AdamMultiGroup(
                [updater.actor_optimizer, updater.alpha_optimizer])

    def fits(self, batch) -> bool:
        return True  # loss kernels grid-stride; no LDS shape limit

    # ------------------------------------------------------------------ #
    def _actor_fwd(self, x, hx0, cx0):
        e = ext()
        mo, _, _, stash = e.seq_lstm_forward(
            x, hx0, cx0, self.actor_core.body_w, self.actor_core.body_b,
            self.actor_core.w_ih, self.actor_core.w_hh, self.actor_core.b_g,
            self.actor_core.heads_w, self.actor_core.heads_b,
        )
        return mo, stash

    def _critic_fwd(self, qmod, obs, act, hx0, cx0):
        """Critic-continuous forward: torch-GEMM encoders + fused core."""
        B, S, _ = obs.shape
        e = ext()
        o = torch.relu(
            obs.reshape(B * S, -1) @ qmod.obs_enc_w + qmod.obs_enc_b)
        a = torch.relu(
            act.reshape(B * S, -1) @ qmod.act_enc_w + qmod.act_enc_b)
        enc = torch.cat([o, a], dim=-1).view(B, S, qmod.hidden_size)
        mo, _, _, stash = e.seq_lstm_forward(
            enc, hx0, cx0, qmod.core.body_w, qmod.core.body_b,
            qmod.core.w_ih, qmod.core.w_hh, qmod.core.b_g,
            qmod.core.heads_w, qmod.core.heads_b,
        )
        return mo, stash, enc, o, a

    def _critic_bwd_dact(self, qmod, gq, stash, enc, cx0):
        """Input-gradient-only critic backward: dminQ/da for the actor."""
        e = ext()
        dx, _, _, _, _ = e.seq_lstm_backward_core(
            gq, None, None, stash, enc, cx0, qmod.core.body_w,
            qmod.core.w_ih, qmod.core.w_hh, qmod.core.heads_w,
        )
        half = qmod.hidden_size // 2
        B, S, _ = enc.shape
        aenc = enc.reshape(B * S, -1)[:, half:]
        dpre_a = dx.reshape(B * S, -1)[:, half:] * (aenc > 0).float()
        return (dpre_a @ qmod.act_enc_w.t()).view(B, S, -1)

    def _critic_bwd_wgrad(self, qmod, gq, stash, enc, obs, act, hx0, cx0,
                          norm):
        """Full critic backward: core wgrads via the MFMA kernels, encoder
        wgrads via library GEMMs written into the flat-grad views."""
        e = ext()
        core = qmod.core
        dx, _, _, dgates, dxb = e.seq_lstm_backward_core(
            gq, None, None, stash, enc, cx0, core.body_w, core.w_ih,
            core.w_hh, core.heads_w,
        )
        gs = [core.body_w.grad, core.body_b.grad, core.w_ih.grad,
              core.w_hh.grad, core.b_g.grad, core.heads_w.grad,
              core.heads_b.grad]
        assert all(g is not None for g in gs)
        e.seq_lstm_wgrad_out(enc, hx0, stash, dgates, dxb, gq,
                             gs[2], gs[3], gs[0], gs[1], gs[4], gs[5], gs[6],
                             norm)
        B, S, _ = enc.shape
        half = qmod.hidden_size // 2
        encf = enc.reshape(B * S, -1)
        dxf = dx.reshape(B * S, -1)
        dpre_o = dxf[:, :half] * (encf[:, :half] > 0).float()
        dpre_a = dxf[:, half:] * (encf[:, half:] > 0).float()
        of = obs.reshape(B * S, -1)
        af = act.reshape(B * S, -1)
        qmod.obs_enc_w.grad.copy_(of.t() @ dpre_o)
        qmod.obs_enc_b.grad.copy_(dpre_o.sum(0))
        qmod.act_enc_w.grad.copy_(af.t() @ dpre_a)
        qmod.act_enc_b.grad.copy_(dpre_a.sum(0))
        if norm is not None:
            norm.add_(qmod.obs_enc_w.grad.pow(2).sum()
                      + qmod.obs_enc_b.grad.pow(2).sum()
                      + qmod.act_enc_w.grad.pow(2).sum()
                      + qmod.act_enc_b.grad.pow(2).sum())

    def _actor_bwd_wgrad(self, gouts, stash, x, hx0, cx0, norm):
        e = ext()
        core = self.actor_core
        _, _, _, dgates, dxb = e.seq_lstm_backward_core(
            gouts, None, None, stash, x, cx0, core.body_w, core.w_ih,
            core.w_hh, core.heads_w,
        )
        gs = [core.body_w.grad, core.body_b.grad, core.w_ih.grad,
              core.w_hh.grad, core.b_g.grad, core.heads_w.grad,
              core.heads_b.grad]
        assert all(g is not None for g in gs)
        e.seq_lstm_wgrad_out(x, hx0, stash, dgates, dxb, gouts,
                             gs[2], gs[3], gs[0], gs[1], gs[4], gs[5], gs[6],
                             norm)

    def _opt(self, optimizer):
        if self.grad_reducer is not None:
            self.grad_reducer.all_reduce([optimizer.flat_grad])
            optimizer.step()
        else:
            optimizer._update()

    # ------------------------------------------------------------------ #
    def _body(self, batch):
        from pdrl_amd.agents.learner_module.compute_loss import soft_update

        u, p = self.u, self.params
        e = ext()
        x = batch["obs"]
        B, S, _ = x.shape
        N = B * S
        A = u.actor.n_outputs
        hx0 = batch["hx"][:, 0]
        cx0 = batch["cx"][:, 0]
        act_b = batch["act"].reshape(B, S, A)
        rew = batch["rew"].reshape(B, S)
        fir = batch["is_fir"].reshape(B, S)
        log_alpha = u.log_alpha.data.view(1)
        single = self.grad_reducer is None
        dev = x.device

        # 1. actor fwd + reparameterized sample
        moA, stA = self._actor_fwd(x, hx0, cx0)
        eps1 = torch.empty(B, S, A, device=dev)
        a1 = torch.empty(B, S, A, device=dev)
        logpi1 = torch.empty(B, S, 1, device=dev)
        e.sacc_sample(moA, self.rng, eps1, a1, logpi1)

        # 2. critics on the fresh sample; cross-network dminQ/da
        qp1, st1p, enc1p, _, _ = self._critic_fwd(self.q["q1"], x, a1, hx0, cx0)
        qp2, st2p, enc2p, _, _ = self._critic_fwd(self.q["q2"], x, a1, hx0, cx0)
        m1 = (qp1 <= qp2).float()
        gq1p = -m1 / N
        gq2p = -(1.0 - m1) / N
        g = self._critic_bwd_dact(self.q["q1"], gq1p, st1p, enc1p, cx0) \
            + self._critic_bwd_dact(self.q["q2"], gq2p, st2p, enc2p, cx0)

        # 3. analytic actor + temperature gradients (+ Adam clock prep)
        dmoA = torch.empty_like(moA)
        clk = u.actor_optimizer if self.adam_aa is not None else None
        e.sacc_actor_grad(
            moA, eps1, a1, g, qp1.reshape(-1), qp2.reshape(-1), log_alpha,
            dmoA, u.log_alpha.grad.view(1), self.stats_buf[:4],
            u.actor_optimizer.norm_sq if single else None,
            None, u.target_entropy,
            clock=clk.state3 if clk is not None else None,
            beta1=clk.beta1 if clk is not None else 0.9,
            beta2=clk.beta2 if clk is not None else 0.999,
        )
        # 4. actor + alpha updates
        self._actor_bwd_wgrad(dmoA, stA, x, hx0, cx0,
                              u.actor_optimizer.norm_sq if single else None)
        if self.adam_aa is not None:
            self.adam_aa.update(tick=False)  # clock prepped by actor_grad
        else:
            self._opt(u.actor_optimizer)
            if self.grad_reducer is not None:
                self.grad_reducer.all_reduce([u.alpha_optimizer.flat_grad])
            u.alpha_optimizer._update()  # no clip: norm unused

        # 5. post-update sample
        moA2, _ = self._actor_fwd(x, hx0, cx0)
        eps2 = torch.empty(B, S, A, device=dev)
        a2 = torch.empty(B, S, A, device=dev)
        logpi2 = torch.empty(B, S, 1, device=dev)
        e.sacc_sample(moA2, self.rng, eps2, a2, logpi2)

        # 6. target critics on a'; behavior critics on the batch actions
        tq1, _, _, _, _ = self._critic_fwd(self.q["t1"], x, a2, hx0, cx0)
        tq2, _, _, _, _ = self._critic_fwd(self.q["t2"], x, a2, hx0, cx0)
        qb1, st1b, enc1b, _, _ = self._critic_fwd(self.q["q1"], x, act_b, hx0, cx0)
        qb2, st2b, enc2b, _, _ = self._critic_fwd(self.q["q2"], x, act_b, hx0, cx0)

        # 7. critic losses
        gq1 = torch.empty_like(qb1)
        gq2 = torch.empty_like(qb2)
        e.sacc_critic_loss(
            qb1.reshape(-1), qb2.reshape(-1), tq1.reshape(-1),
            tq2.reshape(-1), logpi2.reshape(-1), rew, fir, log_alpha,
            gq1.reshape(-1), gq2.reshape(-1), self.stats_buf[4:5],
            u.critic_optimizer.norm_sq if single else None,
            p.gamma, p.reward_scale,
        )
        # 8. critic updates (both stacks accumulate into one flat space)
        cn = u.critic_optimizer.norm_sq if single else None
        self._critic_bwd_wgrad(self.q["q1"], gq1, st1b, enc1b, x, act_b,
                               hx0, cx0, cn)
        self._critic_bwd_wgrad(self.q["q2"], gq2, st2b, enc2b, x, act_b,
                               hx0, cx0, cn)
        self._opt(u.critic_optimizer)

        # 9. Polyak target update
        soft_update(u.critic, u.target_critic, u.TAU)

    def _full(self, batch):
        for _ in range(self.params.K_epoch):
            self._body(batch)
