"""Autograd wiring for the fused SeqLSTMCore HIP kernels.

Forward: ONE kernel computes body GEMM+ReLU, the whole LSTM sequence, and all
heads (concatenated along the output dim), stashing activations for backward.
Backward: ONE kernel runs BPTT through heads/recurrence/body per batch row
(emitting per-step pre-activation gate grads), then the weight gradients run
on the hand-written MFMA kernels (csrc/wgrad.hip: v_mfma_f32_16x16x4_f32
tiles for the gate-weight GEMMs, wave-per-element reductions for the small
grads) over the stashed activations.
"""
from __future__ import annotations

import torch

from . import ext


class _SeqLSTMFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, hx, cx, body_w, body_b, w_ih, w_hh, b_g, heads_w,
                heads_b):
        x = x.contiguous()
        hx = hx.contiguous()
        cx = cx.contiguous()
        outs, hS, cS, stash = ext().seq_lstm_forward(
            x, hx, cx, body_w, body_b, w_ih, w_hh, b_g, heads_w, heads_b,
        )
        ctx.save_for_backward(x, hx, cx, body_w, w_ih, w_hh, heads_w, stash)
        return outs, hS, cS

    @staticmethod
    def backward(ctx, gouts, ghS, gcS):
        x, hx, cx, body_w, w_ih, w_hh, heads_w, stash = ctx.saved_tensors
        gouts = gouts.contiguous()
        ghS_c = ghS.contiguous() if ghS is not None else None
        gcS_c = gcS.contiguous() if gcS is not None else None

        dx, dh0, dc0, dgates, dxb = ext().seq_lstm_backward_core(
            gouts, ghS_c, gcS_c, stash, x, cx, body_w, w_ih, w_hh, heads_w,
        )
        # weight grads: hand-written MFMA GEMMs over the stashed activations
        # (the hprev shift is done by the kernel's addressing — no cat)
        dW_ih, dW_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b = (
            ext().seq_lstm_wgrad(x, hx, stash, dgates, dxb, gouts)
        )
        return (dx, dh0, dc0, dbody_w, dbody_b, dW_ih, dW_hh, db_g, dheads_w,
                dheads_b)


class _SeqLSTMDualFunction(torch.autograd.Function):
    """Dual-body variant: the LSTM input is [relu(x·W1+b1) | relu(x2·W2+b2)]
    (continuous-critic topology). One kernel each way; the encoder weight
    grads ride the same wgrad launch as extra wave-per-element segments."""

    @staticmethod
    def forward(ctx, x, x2, hx, cx, body_w, body_b, body2_w, body2_b, w_ih,
                w_hh, b_g, heads_w, heads_b):
        x = x.contiguous()
        x2 = x2.contiguous()
        hx = hx.contiguous()
        cx = cx.contiguous()
        outs, hS, cS, stash = ext().seq_lstm_forward(
            x, hx, cx, body_w, body_b, w_ih, w_hh, b_g, heads_w, heads_b,
            x2=x2, body2_w=body2_w, body2_b=body2_b,
        )
        ctx.save_for_backward(x, x2, hx, cx, body_w, body2_w, w_ih, w_hh,
                              heads_w, stash)
        return outs, hS, cS

    @staticmethod
    def backward(ctx, gouts, ghS, gcS):
        (x, x2, hx, cx, body_w, body2_w, w_ih, w_hh, heads_w,
         stash) = ctx.saved_tensors
        gouts = gouts.contiguous()
        ghS_c = ghS.contiguous() if ghS is not None else None
        gcS_c = gcS.contiguous() if gcS is not None else None

        dx, dh0, dc0, dgates, dxb, dx2 = ext().seq_lstm_backward_core(
            gouts, ghS_c, gcS_c, stash, x, cx, body_w, w_ih, w_hh, heads_w,
            body2_w=body2_w,
        )
        (dW_ih, dW_hh, dbody_w, dbody_b, db_g, dheads_w, dheads_b, dbody2_w,
         dbody2_b) = ext().seq_lstm_wgrad(x, hx, stash, dgates, dxb, gouts,
                                          x2=x2)
        return (dx, dx2, dh0, dc0, dbody_w, dbody_b, dbody2_w, dbody2_b,
                dW_ih, dW_hh, db_g, dheads_w, dheads_b)


def seq_lstm_apply(core, x, hx, cx, x2=None):
    if x2 is not None:
        outs_cat, hS, cS = _SeqLSTMDualFunction.apply(
            x, x2, hx, cx, core.body_w, core.body_b, core.body2_w,
            core.body2_b, core.w_ih, core.w_hh, core.b_g, core.heads_w,
            core.heads_b,
        )
        return core.split_heads(outs_cat), hS, cS
    outs_cat, hS, cS = _SeqLSTMFunction.apply(
        x, hx, cx, core.body_w, core.body_b, core.w_ih, core.w_hh, core.b_g,
        core.heads_w, core.heads_b,
    )
    return core.split_heads(outs_cat), hS, cS
