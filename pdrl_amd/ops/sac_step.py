"""FusedSacStep: the SAC (discrete) training iteration as a fixed HIP kernel
DAG — the discrete half of K11, hipGraph-capturable.

Default path (_body8, PDRL_SAC8=1): EIGHT launches — launch latency IS the
step time at this size (profiles/algo_breakdown_r02c.md, 103 µs/step):
  1. 5-network forward (actor + twin critics + twin targets)
  2. actor loss (row-local dlogits) + actor BPTT          [sac_actor_bwd]
  3. actor MFMA wgrad + loss reduce + shared Adam clock   [sac_actor_wgrad]
  4. actor + alpha Adam (one multi-group launch)
  5. post-update actor forward + critic loss       [sac_fwd2_critic_loss]
  6. twin-critic BPTT (multi)   7. twin-critic MFMA wgrads (multi)
  8. critic Adam + value-loss reduce + Polyak target (adam_step extras;
     the target critic's params live in a flat buffer aligned with
     critic_optimizer.flat_param)

Legacy 10-launch sequence (_body_legacy, also the multi-rank path),
following the EAGER reference ordering exactly
(agents/learner_module/sac/learning.py):
  1.  actor + twin critics + twin TARGET critics — ONE 5-network launch
      (target params are constant until step 9, so their forward commutes
      with the actor update)
  2.  sac_actor_loss      — analytic dlogits + dlog_alpha + stats
                            (+ the shared Adam clock prep, in-kernel)
  3.  actor BPTT + MFMA wgrad → flat actor grads (+ norm)
  4.  actor/alpha Adam — ONE multi-group launch
  5.  actor fwd AGAIN (post-update policy, as the reference does)
  7.  sac_critic_loss     — soft-Q TD target, twin huber, analytic dq1/dq2
  8.  q1+q2 BPTT (one launch) + q1+q2 MFMA wgrads (one launch) → Adam
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
        import os
        # 8-launch restructured DAG (H=64, single-rank): actor loss rides
        # the actor BPTT, its reduce + Adam clock ride the actor wgrad,
        # the post-update forward rides the critic loss, and the vl reduce
        # + Polyak ride the critic Adam. PDRL_SAC8=0 falls back to the
        # 10-launch DAG (kept for multi-rank and other widths).
        self._fast8 = (self.grad_reducer is None
                       and self.cores["actor"].w_ih.size(0) == 64
                       and bool(int(os.environ.get("PDRL_SAC8", "1"))))
        # actor+alpha Adam updates happen back-to-back at the same DAG
        # point → one multi-group launch (shared device step clock)
        self.adam_aa = None
        if self.grad_reducer is None and getattr(
                updater.actor_optimizer, "shared_clock", False):
            from .optim import AdamMultiGroup

            self.adam_aa = AdamMultiGroup(
                [updater.actor_optimizer, updater.alpha_optimizer])

    def fits(self, batch) -> bool:
        B, S, _ = batch["obs"].shape
        # wgrad kernels stage a (B*S)-entry row-pointer table in LDS
        return B * S <= 8192

    # ------------------------------------------------------------------ #
    def _multi_setup(self, batch):
        """Preallocate per-core output buffers and build the device pointer
        tables for the multi-network launches (all five cores share one
        head width here: logits(A) for the actor, Q(A) for the critics).
        Tables reference persistent storage only (parameters/grads are
        views into the flat optimizer buffers; activations below are owned
        here), so graph capture replays them safely."""
        c = self.cores
        u = self.u
        x = batch["obs"]
        B, S, _ = x.shape
        dev = x.device
        H = c["actor"].w_ih.size(0)
        D = c["actor"].heads_w.size(1)
        assert all(cc.heads_w.size(1) == D for cc in c.values())
        self._D = D

        def mk(*shape):
            return torch.empty(*shape, device=dev)

        # Re-home the target critic's params into ONE flat fp32 buffer laid
        # out exactly like critic_optimizer.flat_param (same param order —
        # target is a deepcopy), so the Polyak update can ride the critic
        # Adam kernel. Must precede the pointer-table builds below (p.data
        # storage moves). Params stay views, so checkpoint/load still work.
        if not hasattr(self, "_t_flat"):
            cps = u.critic_optimizer.space.params
            tps = list(u.target_critic.parameters())
            assert len(cps) == len(tps)
            tf = torch.empty(sum(p.numel() for p in tps), device=dev)
            off = 0
            for cp, tp in zip(cps, tps):
                assert cp.shape == tp.shape, "critic/target param order skew"
                n = tp.numel()
                tf[off:off + n].copy_(tp.data.reshape(-1))
                tp.data = tf[off:off + n].view_as(tp.data)
                off += n
            self._t_flat = tf

        buf = {}
        for name in ("actor", "q1", "q2", "t1", "t2"):
            buf[name] = {"outs": mk(B, S, D), "hS": mk(B, H), "cS": mk(B, H),
                         "stash": mk(B, S, 7 * H)}
        for name in ("q1", "q2"):
            buf[name]["dgates"] = mk(B, S, 4 * H)
            buf[name]["dxb"] = mk(B, S, H)
            buf[name]["gq"] = mk(B, S, D)
        if self._fast8:
            a = buf["actor"]
            a["gA"] = mk(B, S, D)
            a["dgates"] = mk(B, S, 4 * H)
            a["dxb"] = mk(B, S, H)
            buf["mo2"] = mk(B, S, D)
            self._sp_a = mk(B, 2)   # actor-loss {ubar, entropy} partials
            self._sp_c = mk(B)      # critic huber partials
            ag = self.cores["actor"]
            gs = [ag.body_w.grad, ag.body_b.grad, ag.w_ih.grad,
                  ag.w_hh.grad, ag.b_g.grad, ag.heads_w.grad,
                  ag.heads_b.grad]
            assert all(g is not None for g in gs)
            self._actor_grads = gs
        self.buf = buf

        def t64(rows):
            return torch.tensor(rows, dtype=torch.int64).to(dev)

        def wrow(core):
            # 10-wide row; last three (body2_w, body2_b, x2) are the
            # dual-body slots, unused for the discrete (single-body) cores
            return [core.body_w.data_ptr(), core.body_b.data_ptr(),
                    core.w_ih.data_ptr(), core.w_hh.data_ptr(),
                    core.b_g.data_ptr(), core.heads_w.data_ptr(),
                    core.heads_b.data_ptr(), 0, 0, 0]

        def orow(b):
            return [b["outs"].data_ptr(), b["hS"].data_ptr(),
                    b["cS"].data_ptr(), b["stash"].data_ptr()]

        # fwd1 carries FIVE networks: actor + twin critics + twin TARGET
        # critics — the targets' parameters only move at the Polyak update
        # at the END of the step, so their forward commutes with the actor
        # update and joins the first launch. Only the post-update actor
        # re-forward remains separate.
        self.fwd1_cores = t64([wrow(c["actor"]), wrow(c["q1"]), wrow(c["q2"]),
                               wrow(c["t1"]), wrow(c["t2"])])
        self.fwd1_outs = t64([orow(buf["actor"]), orow(buf["q1"]),
                              orow(buf["q2"]), orow(buf["t1"]),
                              orow(buf["t2"])])

        def birow(name):
            core, b = c[name], buf[name]
            return [b["gq"].data_ptr(), b["stash"].data_ptr(),
                    core.w_ih.data_ptr(), core.w_hh.data_ptr(),
                    core.heads_w.data_ptr(), core.body_w.data_ptr(), 0]

        def borow(name):
            b = buf[name]
            return [b["dgates"].data_ptr(), b["dxb"].data_ptr(), 0]

        self.bwd_in = t64([birow("q1"), birow("q2")])
        self.bwd_out = t64([borow("q1"), borow("q2")])

        nrm = (u.critic_optimizer.norm_sq.data_ptr()
               if self.grad_reducer is None else 0)

        def grow(name):
            core, b = c[name], buf[name]
            g = [core.body_w.grad, core.body_b.grad, core.w_ih.grad,
                 core.w_hh.grad, core.b_g.grad, core.heads_w.grad,
                 core.heads_b.grad]
            assert all(t is not None for t in g)
            return [b["stash"].data_ptr(), b["dgates"].data_ptr(),
                    b["dxb"].data_ptr(), b["gq"].data_ptr(),
                    g[2].data_ptr(), g[3].data_ptr(), g[0].data_ptr(),
                    g[1].data_ptr(), g[4].data_ptr(), g[5].data_ptr(),
                    g[6].data_ptr(), nrm, 0, 0]

        self.wg_tab = t64([grow("q1"), grow("q2")])
        self._mshape = (B, S)

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
        if self._fast8:
            return self._body8(batch)
        return self._body_legacy(batch)

    def _body8(self, batch):
        """8-launch restructured DAG (see __init__); math identical to the
        legacy sequence (GPU parity test vs the eager updater)."""
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
        if getattr(self, "_mshape", None) != (B, S):
            self._multi_setup(batch)
        buf, D = self.buf, self._D
        ac = self.cores["actor"]
        ab = buf["actor"]

        # 1. actor + twin critics + twin TARGET critics — ONE launch
        e.seq_lstm_forward_multi(x, hx0, cx0, self.fwd1_cores,
                                 self.fwd1_outs, 5, D)
        moA1 = ab["outs"]
        mq1, mq2 = buf["q1"]["outs"], buf["q2"]["outs"]

        # 2. actor loss (row-local analytic dlogits) + actor BPTT — ONE
        #    launch; per-row loss partials to _sp_a; zeroes the actor norm
        e.sac_actor_bwd(moA1, mq1, mq2, log_alpha, ab["gA"], self._sp_a,
                        u.actor_optimizer.norm_sq, ab["stash"], x, cx0,
                        ac.body_w, ac.w_ih, ac.w_hh, ac.heads_w,
                        ab["dgates"], ab["dxb"])
        # 3. actor MFMA wgrad + loss reduce (g_alpha, stats[0..3]) + shared
        #    Adam clock prep — ONE launch (one extra block on the wgrad grid)
        gs = self._actor_grads
        clk = u.actor_optimizer if self.adam_aa is not None else None
        e.sac_actor_wgrad(
            x, hx0, ab["stash"], ab["dgates"], ab["dxb"], ab["gA"],
            gs[2], gs[3], gs[0], gs[1], gs[4], gs[5], gs[6],
            u.actor_optimizer.norm_sq, self._sp_a, log_alpha,
            u.log_alpha.grad.view(1), self.stats_buf[:4], alpha_norm=None,
            clock=clk.state3 if clk is not None else None,
            target_entropy=u.target_entropy,
            beta1=clk.beta1 if clk is not None else 0.9,
            beta2=clk.beta2 if clk is not None else 0.999,
        )
        # 4. actor + alpha Adam — ONE multi-group launch
        if self.adam_aa is not None:
            self.adam_aa.update(tick=False)  # clock prepped by the wgrad
        else:
            u.actor_optimizer._update()
            u.alpha_optimizer._update()  # no clip: norm unused

        # 5. post-update actor forward + critic loss — ONE launch (the
        #    actor stash/hS/cS scratch is dead after step 3 and is reused;
        #    zeroes the critic norm; vl partials to _sp_c)
        gq1, gq2 = buf["q1"]["gq"], buf["q2"]["gq"]
        e.sac_fwd2_critic_loss(
            x, hx0, cx0, ac.body_w, ac.body_b, ac.w_ih, ac.w_hh, ac.b_g,
            ac.heads_w, ac.heads_b, buf["mo2"], ab["hS"], ab["cS"],
            ab["stash"], mq1, mq2, buf["t1"]["outs"], buf["t2"]["outs"],
            act, rew, fir, log_alpha, gq1, gq2, self._sp_c,
            u.critic_optimizer.norm_sq, p.gamma, p.reward_scale,
        )
        # 6-7. twin-critic BPTT + MFMA wgrads — one launch each
        e.seq_lstm_backward_multi(x, cx0, self.bwd_in, self.bwd_out, 2, D)
        e.seq_lstm_wgrad_multi(x, hx0, self.wg_tab, 2, D)
        # 8. critic Adam + vl-stat reduce + Polyak target — ONE launch
        co = u.critic_optimizer
        e.adam_step(
            co.space.flat_param, co.space.flat_grad, co.exp_avg,
            co.exp_avg_sq, co.state3, co.norm_sq, co.lr, co.beta1, co.beta2,
            co.eps, co.max_norm, do_prep=False, stats_part=self._sp_c,
            stats_out=self.stats_buf[4:5], part_scale=1.0 / (B * (S - 1)),
            polyak=self._t_flat, tau=u.TAU,
        )

    def _body_legacy(self, batch):
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
        if getattr(self, "_mshape", None) != (B, S):
            self._multi_setup(batch)
        buf, D = self.buf, self._D

        # 1. actor + twin critics + twin TARGET critics — ONE 5-network
        #    launch (the targets' params are constant until the Polyak
        #    update at step 9, so their forward commutes with steps 2-5)
        e.seq_lstm_forward_multi(x, hx0, cx0, self.fwd1_cores,
                                 self.fwd1_outs, 5, D)
        moA1 = buf["actor"]["outs"]
        mq1, mq2 = buf["q1"]["outs"], buf["q2"]["outs"]

        # 2. actor + temperature losses (analytic grads); the shared Adam
        #    clock is prepped inside this kernel (one less launch)
        gA = torch.empty_like(moA1)
        clk = u.actor_optimizer if self.adam_aa is not None else None
        e.sac_actor_loss(
            moA1, mq1, mq2, log_alpha, gA, u.log_alpha.grad.view(1),
            self.stats_buf[:4], u.actor_optimizer.norm_sq if single else None,
            None, u.target_entropy,
            clock=clk.state3 if clk is not None else None,
            beta1=clk.beta1 if clk is not None else 0.9,
            beta2=clk.beta2 if clk is not None else 0.999,
        )
        # 3-4. actor + alpha updates
        self._bwd_wgrad(self.cores["actor"], gA, buf["actor"]["stash"], x,
                        hx0, cx0,
                        u.actor_optimizer.norm_sq if single else None)
        if self.adam_aa is not None:
            self.adam_aa.update(tick=False)  # clock prepped by the loss kernel
        else:
            self._opt(u.actor_optimizer)
            if self.grad_reducer is not None:
                self.grad_reducer.all_reduce([u.alpha_optimizer.flat_grad])
            u.alpha_optimizer._update()  # no clip: norm unused

        # 5. post-update actor forward (single network; the targets were
        #    already evaluated in the 5-network launch)
        ac = self.cores["actor"]
        moA2, _, _, _ = e.seq_lstm_forward(
            x, hx0, cx0, ac.body_w, ac.body_b, ac.w_ih, ac.w_hh, ac.b_g,
            ac.heads_w, ac.heads_b,
        )
        mt1, mt2 = buf["t1"]["outs"], buf["t2"]["outs"]

        # 7. critic losses (head grads land in the persistent gq buffers
        #    the bwd/wgrad pointer tables reference)
        gq1, gq2 = buf["q1"]["gq"], buf["q2"]["gq"]
        e.sac_critic_loss(
            moA2, mq1, mq2, mt1, mt2, act, rew, fir, log_alpha, gq1, gq2,
            self.stats_buf[4:5], u.critic_optimizer.norm_sq if single else None,
            p.gamma, p.reward_scale,
        )
        # 8. twin-critic backward + MFMA weight grads — one launch each
        e.seq_lstm_backward_multi(x, cx0, self.bwd_in, self.bwd_out, 2, D)
        e.seq_lstm_wgrad_multi(x, hx0, self.wg_tab, 2, D)
        self._opt(u.critic_optimizer)

        # 9. Polyak target update
        soft_update(u.critic, u.target_critic, u.TAU)

    def _full(self, batch):
        for _ in range(self.params.K_epoch):
            self._body(batch)
