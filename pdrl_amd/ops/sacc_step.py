"""FusedSacContinuousStep: the SAC-Continuous training iteration as a fixed
HIP kernel DAG — the continuous half of K11,
hipGraph-capturable.

Default path (_body11, PDRL_SAC8=1): ELEVEN launches
(profiles/algo_breakdown_r02c.md, 137 µs/step) — sampling rides the actor
forwards (+ dQ/da zeroing / behaviour-action staging), the min-critic
masks ride the twin input-grad backward, the analytic actor grad rides
the actor BPTT (cross-row reduce + Adam clock on the wgrad extra block,
reusing the discrete reduce verbatim), the soft-Q critic loss rides the
twin-critic BPTT, and the value-loss reduce + Polyak ride the critic
Adam. (Round-1 eager version ran ~95 launches; the first fused DAG 18.)

Legacy 18-launch sequence (_body_legacy, also the multi-rank path),
following the EAGER reference ordering exactly
(agents/learner_module/sac_continuous/learning.py, reference
sac_continuous/learning.py:13-151):
  1.  actor fwd → reparameterized tanh-Gaussian sample (sacc_sample:
      in-kernel counter RNG, graph-replay safe) → a1, log pi
  2.  twin critics on (obs, a1) — ONE dual-body multi-network launch (the
      critic is a dual-body SeqLSTMCore: obs/action encoders feed the LSTM
      directly, matching the reference topology)
  3.  sacc_min_mask — min-critic selection grads + zero the dQ/da buffer
  4.  twin-critic input-grad backward — ONE multi launch, both critics
      atomically accumulate dminQ/da into the shared buffer
  5.  sacc_actor_grad — analytic dmu/dlog_std + dlog_alpha + stats + the
      shared Adam clock prep (all in-kernel)
  6.  actor BPTT + MFMA wgrad → actor+alpha Adam in ONE multi-group launch
  7.  actor fwd AGAIN (post-update sample a2, log pi2)
  8.  target critics on (obs,a2) + behavior critics on (obs,act) — ONE
      4-network dual-body launch (per-core x2 pointers)
  9.  sacc_critic_loss — soft-Q target, twin huber, analytic dq1/dq2
  10. twin-critic BPTT (one launch) + MFMA wgrads incl. encoder segments
      (one launch) → critic Adam
  11. Polyak soft_update (cached device pointer tables)
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
            "q1": updater.critic.q1.core,
            "q2": updater.critic.q2.core,
            "t1": updater.target_critic.q1.core,
            "t2": updater.target_critic.q2.core,
        }
        dev = self.actor_core.body_w.device
        self.stats_buf = torch.zeros(8, dtype=torch.float32, device=dev)
        self.stat_names = _SACC_STATS
        self.rng = torch.randint(1, 1 << 30, (1,), dtype=torch.int32, device=dev)
        self.use_graph = use_graph and self.grad_reducer is None
        import os
        # 11-launch restructured DAG (H=64, single-rank) mirroring the
        # SAC-discrete restructure: sampling rides the actor forwards,
        # min-mask rides the input-grad backward, actor grad rides the
        # actor BPTT (reduce + clock on the wgrad extra block), critic
        # loss rides the critic BPTT, vl-reduce + Polyak ride the critic
        # Adam. PDRL_SAC8=0 falls back to the 18-launch DAG.
        self._fast11 = (self.grad_reducer is None
                        and self.actor_core.w_ih.size(0) == 64
                        and bool(int(os.environ.get("PDRL_SAC8", "1"))))
        # actor+alpha Adam updates batch into one launch (shared clock)
        self.adam_aa = None
        if self.grad_reducer is None and getattr(
                updater.actor_optimizer, "shared_clock", False):
            from .optim import AdamMultiGroup

            self.adam_aa = AdamMultiGroup(
                [updater.actor_optimizer, updater.alpha_optimizer])
        self._mshape = None

    def fits(self, batch) -> bool:
        B, S, _ = batch["obs"].shape
        # wgrad kernels stage a (B*S)-entry row-pointer table in LDS
        return B * S <= 8192

    # ------------------------------------------------------------------ #
    def _multi_setup(self, batch):
        """Preallocate activation/grad buffers and device pointer tables for
        the multi-network dual-body launches. Tables reference persistent
        storage only (graph-replay safe): parameters/grads are flat-space
        views; a1/a2/actb sample buffers are owned here."""
        u = self.u
        x = batch["obs"]
        B, S, _ = x.shape
        dev = x.device
        ac = self.actor_core
        H = ac.w_ih.size(0)
        A = u.actor.n_outputs
        self._A = A
        q1 = self.q["q1"]
        assert q1.body2_w is not None, "continuous critic must be dual-body"
        self._half = q1.body_w.size(1)

        def mk(*shape):
            return torch.empty(*shape, device=dev)

        # Re-home the target critic's params into ONE flat buffer laid out
        # like critic_optimizer.flat_param (see sac_step.py) so the Polyak
        # update can ride the critic Adam kernel. Must precede the pointer
        # tables below (p.data storage moves).
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

        buf = {
            "a1": mk(B, S, A), "eps1": mk(B, S, A), "logpi1": mk(B, S, 1),
            "a2": mk(B, S, A), "eps2": mk(B, S, A), "logpi2": mk(B, S, 1),
            "actb": mk(B, S, A),
            "dact": mk(B, S, A),           # accumulated dminQ/da
            "gq1p": mk(B, S, 1), "gq2p": mk(B, S, 1),
        }
        if self._fast11:
            buf["moA"] = mk(B, S, 2 * A)
            buf["moA2"] = mk(B, S, 2 * A)
            buf["dmoA"] = mk(B, S, 2 * A)
            buf["stA"] = mk(B, S, 7 * H)
            buf["hA"] = mk(B, H)
            buf["cA"] = mk(B, H)
            buf["adg"] = mk(B, S, 4 * H)
            buf["adx"] = mk(B, S, H)
            self._sp_a = mk(B, 2)   # actor partials {l_sum, -lp_sum}
            self._sp_c = mk(B, 2)   # per-(row, critic) huber partials
            ag = ac
            gs = [ag.body_w.grad, ag.body_b.grad, ag.w_ih.grad,
                  ag.w_hh.grad, ag.b_g.grad, ag.heads_w.grad,
                  ag.heads_b.grad]
            assert all(g is not None for g in gs)
            self._actor_grads = gs
        # per-network activation slots (critics D=1)
        for name in ("q1s", "q2s", "t1", "t2", "q1b", "q2b"):
            buf[name] = {"outs": mk(B, S, 1), "hS": mk(B, H), "cS": mk(B, H),
                         "stash": mk(B, S, 7 * H)}
        for name in ("q1s", "q2s", "q1b", "q2b"):
            buf[name]["dgates"] = mk(B, S, 4 * H)
            buf[name]["dxb"] = mk(B, S, H)
        for name in ("q1b", "q2b"):
            buf[name]["gq"] = mk(B, S, 1)
        self.buf = buf

        def t64(rows):
            return torch.tensor(rows, dtype=torch.int64).to(dev)

        def wrow(core, x2_buf):
            return [core.body_w.data_ptr(), core.body_b.data_ptr(),
                    core.w_ih.data_ptr(), core.w_hh.data_ptr(),
                    core.b_g.data_ptr(), core.heads_w.data_ptr(),
                    core.heads_b.data_ptr(), core.body2_w.data_ptr(),
                    core.body2_b.data_ptr(), x2_buf.data_ptr()]

        def orow(b):
            return [b["outs"].data_ptr(), b["hS"].data_ptr(),
                    b["cS"].data_ptr(), b["stash"].data_ptr()]

        q = self.q
        # sample-pass critics on (x, a1)
        self.fwdS_cores = t64([wrow(q["q1"], buf["a1"]),
                               wrow(q["q2"], buf["a1"])])
        self.fwdS_outs = t64([orow(buf["q1s"]), orow(buf["q2s"])])
        # target critics on (x, a2) + behavior critics on (x, actb)
        self.fwdT_cores = t64([
            wrow(q["t1"], buf["a2"]), wrow(q["t2"], buf["a2"]),
            wrow(q["q1"], buf["actb"]), wrow(q["q2"], buf["actb"]),
        ])
        self.fwdT_outs = t64([orow(buf["t1"]), orow(buf["t2"]),
                              orow(buf["q1b"]), orow(buf["q2b"])])

        def birow(core, gq, b, want_dgates=True):
            return [gq.data_ptr(), b["stash"].data_ptr(),
                    core.w_ih.data_ptr(), core.w_hh.data_ptr(),
                    core.heads_w.data_ptr(), core.body_w.data_ptr(),
                    core.body2_w.data_ptr()]

        # input-grad pass: dminQ/da accumulated into dact (dgates skipped)
        self.bwdP_in = t64([birow(q["q1"], buf["gq1p"], buf["q1s"]),
                            birow(q["q2"], buf["gq2p"], buf["q2s"])])
        self.bwdP_out = t64([
            [buf["q1s"]["dgates"].data_ptr(), buf["q1s"]["dxb"].data_ptr(),
             buf["dact"].data_ptr()],
            [buf["q2s"]["dgates"].data_ptr(), buf["q2s"]["dxb"].data_ptr(),
             buf["dact"].data_ptr()],
        ])
        # loss pass: dgates/dxb for the weight grads, no input grads
        self.bwdB_in = t64([birow(q["q1"], buf["q1b"]["gq"], buf["q1b"]),
                            birow(q["q2"], buf["q2b"]["gq"], buf["q2b"])])
        self.bwdB_out = t64([
            [buf["q1b"]["dgates"].data_ptr(), buf["q1b"]["dxb"].data_ptr(), 0],
            [buf["q2b"]["dgates"].data_ptr(), buf["q2b"]["dxb"].data_ptr(), 0],
        ])

        nrm = (u.critic_optimizer.norm_sq.data_ptr()
               if self.grad_reducer is None else 0)

        def grow(core, b):
            g = [core.body_w.grad, core.body_b.grad, core.w_ih.grad,
                 core.w_hh.grad, core.b_g.grad, core.heads_w.grad,
                 core.heads_b.grad, core.body2_w.grad, core.body2_b.grad]
            assert all(t is not None for t in g)
            return [b["stash"].data_ptr(), b["dgates"].data_ptr(),
                    b["dxb"].data_ptr(), b["gq"].data_ptr(),
                    g[2].data_ptr(), g[3].data_ptr(), g[0].data_ptr(),
                    g[1].data_ptr(), g[4].data_ptr(), g[5].data_ptr(),
                    g[6].data_ptr(), nrm, g[7].data_ptr(), g[8].data_ptr()]

        self.wgB_tab = t64([grow(q["q1"], buf["q1b"]),
                            grow(q["q2"], buf["q2b"])])
        self._mshape = (B, S)

    def _actor_fwd(self, x, hx0, cx0):
        e = ext()
        c = self.actor_core
        mo, _, _, stash = e.seq_lstm_forward(
            x, hx0, cx0, c.body_w, c.body_b, c.w_ih, c.w_hh, c.b_g,
            c.heads_w, c.heads_b,
        )
        return mo, stash

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
        if self._fast11:
            return self._body11(batch)
        return self._body_legacy(batch)

    def _body11(self, batch):
        """11-launch restructured DAG (see __init__); math identical to the
        legacy sequence (GPU parity tests vs eager)."""
        u, p = self.u, self.params
        e = ext()
        x = batch["obs"]
        B, S, _ = x.shape
        if self._mshape != (B, S):
            self._multi_setup(batch)
        buf = self.buf
        A = self._A
        half = self._half
        hx0 = batch["hx"][:, 0]
        cx0 = batch["cx"][:, 0]
        rew = batch["rew"].reshape(B, S)
        fir = batch["is_fir"].reshape(B, S)
        log_alpha = u.log_alpha.data.view(1)
        ac = self.actor_core
        q = self.q

        # 1. actor fwd + reparameterized sample + zero the dQ/da buffer
        e.sacc_fwd_sample(x, hx0, cx0, ac.body_w, ac.body_b, ac.w_ih,
                          ac.w_hh, ac.b_g, ac.heads_w, ac.heads_b,
                          buf["moA"], buf["hA"], buf["cA"], buf["stA"],
                          self.rng, buf["eps1"], buf["a1"], buf["logpi1"],
                          dact_zero=buf["dact"])
        # 2. twin critics on the fresh sample — ONE dual-body launch
        e.seq_lstm_forward_multi(x, hx0, cx0, self.fwdS_cores, self.fwdS_outs,
                                 2, 1, F2=A, half=half)
        qp1 = buf["q1s"]["outs"]
        qp2 = buf["q2s"]["outs"]
        # 3. min-critic selection + input-grad backward — ONE launch,
        #    dminQ/da atomically accumulated into dact
        e.sacc_minmask_bwd(
            qp1.reshape(-1), qp2.reshape(-1), buf["gq1p"], buf["gq2p"],
            buf["q1s"]["stash"], buf["q2s"]["stash"], x, cx0,
            q["q1"].w_ih, q["q1"].w_hh, q["q1"].heads_w, q["q1"].body2_w,
            q["q2"].w_ih, q["q2"].w_hh, q["q2"].heads_w, q["q2"].body2_w,
            buf["q1s"]["dgates"], buf["q1s"]["dxb"],
            buf["q2s"]["dgates"], buf["q2s"]["dxb"], buf["dact"], half)
        # 4. analytic actor grad (row-local) + actor BPTT — ONE launch
        e.sacc_actor_bwd(
            buf["moA"], buf["eps1"], buf["a1"], buf["dact"],
            qp1.reshape(-1), qp2.reshape(-1), log_alpha, buf["dmoA"],
            self._sp_a, u.actor_optimizer.norm_sq, buf["stA"], x, cx0,
            ac.w_ih, ac.w_hh, ac.heads_w, buf["adg"], buf["adx"])
        # 5. actor wgrad + loss reduce + shared Adam clock prep (the
        #    continuous partials are stored in the discrete form, so the
        #    discrete reduce block applies verbatim)
        gs = self._actor_grads
        clk = u.actor_optimizer if self.adam_aa is not None else None
        e.sac_actor_wgrad(
            x, hx0, buf["stA"], buf["adg"], buf["adx"], buf["dmoA"],
            gs[2], gs[3], gs[0], gs[1], gs[4], gs[5], gs[6],
            u.actor_optimizer.norm_sq, self._sp_a, log_alpha,
            u.log_alpha.grad.view(1), self.stats_buf[:4], alpha_norm=None,
            clock=clk.state3 if clk is not None else None,
            target_entropy=u.target_entropy,
            beta1=clk.beta1 if clk is not None else 0.9,
            beta2=clk.beta2 if clk is not None else 0.999,
        )
        # 6. actor + alpha Adam — ONE multi-group launch
        if self.adam_aa is not None:
            self.adam_aa.update(tick=False)
        else:
            u.actor_optimizer._update()
            u.alpha_optimizer._update()
        # 7. post-update actor fwd + sample + behaviour-action staging
        e.sacc_fwd_sample(x, hx0, cx0, ac.body_w, ac.body_b, ac.w_ih,
                          ac.w_hh, ac.b_g, ac.heads_w, ac.heads_b,
                          buf["moA2"], buf["hA"], buf["cA"], buf["stA"],
                          self.rng, buf["eps2"], buf["a2"], buf["logpi2"],
                          act_src=batch["act"].reshape(B, S, A).contiguous(),
                          actb=buf["actb"])
        # 8. target critics on a2 + behaviour critics — ONE 4-network launch
        e.seq_lstm_forward_multi(x, hx0, cx0, self.fwdT_cores, self.fwdT_outs,
                                 4, 1, F2=A, half=half)
        # 9. critic loss (row-local) + twin-critic BPTT — ONE launch
        e.sacc_critic_bwd(
            buf["q1b"]["outs"].reshape(-1), buf["q2b"]["outs"].reshape(-1),
            buf["t1"]["outs"].reshape(-1), buf["t2"]["outs"].reshape(-1),
            buf["logpi2"].reshape(-1), rew, fir, log_alpha,
            buf["q1b"]["gq"].reshape(-1), buf["q2b"]["gq"].reshape(-1),
            self._sp_c, u.critic_optimizer.norm_sq,
            buf["q1b"]["stash"], buf["q2b"]["stash"], x, cx0,
            q["q1"].w_ih, q["q1"].w_hh, q["q1"].heads_w,
            q["q2"].w_ih, q["q2"].w_hh, q["q2"].heads_w,
            buf["q1b"]["dgates"], buf["q1b"]["dxb"],
            buf["q2b"]["dgates"], buf["q2b"]["dxb"],
            p.gamma, p.reward_scale)
        # 10. twin-critic MFMA wgrads (encoder grads ride the same launch)
        e.seq_lstm_wgrad_multi(x, hx0, self.wgB_tab, 2, 1, x2=buf["actb"],
                               F2=A, half=half)
        # 11. critic Adam + vl-stat reduce + Polyak target — ONE launch
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
        if self._mshape != (B, S):
            self._multi_setup(batch)
        buf = self.buf
        A = self._A
        half = self._half
        hx0 = batch["hx"][:, 0]
        cx0 = batch["cx"][:, 0]
        rew = batch["rew"].reshape(B, S)
        fir = batch["is_fir"].reshape(B, S)
        log_alpha = u.log_alpha.data.view(1)
        single = self.grad_reducer is None

        # 1. actor fwd + reparameterized sample
        moA, stA = self._actor_fwd(x, hx0, cx0)
        e.sacc_sample(moA, self.rng, buf["eps1"], buf["a1"], buf["logpi1"])

        # 2. twin critics on the fresh sample — ONE dual-body launch
        e.seq_lstm_forward_multi(x, hx0, cx0, self.fwdS_cores, self.fwdS_outs,
                                 2, 1, F2=A, half=half)
        qp1 = buf["q1s"]["outs"]
        qp2 = buf["q2s"]["outs"]

        # 3. min-critic selection grads + zero the dQ/da accumulator
        e.sacc_min_mask(qp1, qp2, buf["gq1p"], buf["gq2p"], buf["dact"])

        # 4. cross-network dminQ/da — ONE multi launch, atomic accumulate
        e.seq_lstm_backward_multi(x, cx0, self.bwdP_in, self.bwdP_out, 2, 1,
                                  F2=A, half=half, accum_dx2=True)

        # 5. analytic actor + temperature gradients (+ Adam clock prep)
        dmoA = torch.empty_like(moA)
        clk = u.actor_optimizer if self.adam_aa is not None else None
        e.sacc_actor_grad(
            moA, buf["eps1"], buf["a1"], buf["dact"], qp1.reshape(-1),
            qp2.reshape(-1), log_alpha, dmoA, u.log_alpha.grad.view(1),
            self.stats_buf[:4],
            u.actor_optimizer.norm_sq if single else None,
            None, u.target_entropy,
            clock=clk.state3 if clk is not None else None,
            beta1=clk.beta1 if clk is not None else 0.9,
            beta2=clk.beta2 if clk is not None else 0.999,
        )
        # 6. actor + alpha updates
        self._actor_bwd_wgrad(dmoA, stA, x, hx0, cx0,
                              u.actor_optimizer.norm_sq if single else None)
        if self.adam_aa is not None:
            self.adam_aa.update(tick=False)  # clock prepped by actor_grad
        else:
            self._opt(u.actor_optimizer)
            if self.grad_reducer is not None:
                self.grad_reducer.all_reduce([u.alpha_optimizer.flat_grad])
            u.alpha_optimizer._update()  # no clip: norm unused

        # 7. post-update sample
        moA2, _ = self._actor_fwd(x, hx0, cx0)
        e.sacc_sample(moA2, self.rng, buf["eps2"], buf["a2"], buf["logpi2"])

        # 8. target critics on a2 + behavior critics on the batch actions —
        #    ONE 4-network launch (per-core x2 pointers in the table; the
        #    batch actions are copied into the persistent actb buffer the
        #    table references, so non-graph callers stay correct)
        buf["actb"].copy_(batch["act"].reshape(B, S, A), non_blocking=True)
        e.seq_lstm_forward_multi(x, hx0, cx0, self.fwdT_cores, self.fwdT_outs,
                                 4, 1, F2=A, half=half)

        # 9. critic losses (head grads land in the persistent gq buffers)
        e.sacc_critic_loss(
            buf["q1b"]["outs"].reshape(-1), buf["q2b"]["outs"].reshape(-1),
            buf["t1"]["outs"].reshape(-1), buf["t2"]["outs"].reshape(-1),
            buf["logpi2"].reshape(-1), rew, fir, log_alpha,
            buf["q1b"]["gq"].reshape(-1), buf["q2b"]["gq"].reshape(-1),
            self.stats_buf[4:5],
            u.critic_optimizer.norm_sq if single else None,
            p.gamma, p.reward_scale,
        )
        # 10. twin-critic backward + MFMA wgrads (encoder grads ride the
        #     same launch as dual-body segments)
        e.seq_lstm_backward_multi(x, cx0, self.bwdB_in, self.bwdB_out, 2, 1,
                                  F2=A, half=half)
        e.seq_lstm_wgrad_multi(x, hx0, self.wgB_tab, 2, 1, x2=buf["actb"],
                               F2=A, half=half)
        self._opt(u.critic_optimizer)

        # 11. Polyak target update
        soft_update(u.critic, u.target_critic, u.TAU)

    def _full(self, batch):
        for _ in range(self.params.K_epoch):
            self._body(batch)
