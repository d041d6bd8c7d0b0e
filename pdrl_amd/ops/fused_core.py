"""Autograd wiring for the fused SeqLSTMCore HIP kernels (forward+backward
of body GEMM + LSTM recurrence + heads in single launches — kernels K1-K3 in
SURVEY.md §2.4). Implemented alongside the HIP extension; see
pdrl_amd/ops/csrc/.
"""
from __future__ import annotations

import torch

from . import ext


class _SeqLSTMFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, core, x, hx, cx, *params):
        head_ws = [core.head_params(n)[0] for n in core.head_names]
        head_bs = [core.head_params(n)[1] for n in core.head_names]
        outs = ext().seq_lstm_forward(
            x, hx, cx,
            core.body_w, core.body_b, core.w_ih, core.w_hh, core.b_g,
            head_ws, head_bs,
        )
        # outs: [head outputs...] + [h_S, c_S, stash]
        n_heads = len(core.head_names)
        heads = outs[:n_heads]
        h_S, c_S, stash = outs[n_heads], outs[n_heads + 1], outs[n_heads + 2]
        ctx.save_for_backward(
            x, hx, cx, core.body_w, core.body_b, core.w_ih, core.w_hh, core.b_g,
            *head_ws, *head_bs, stash,
        )
        ctx.n_heads = n_heads
        return (*heads, h_S, c_S)

    @staticmethod
    def backward(ctx, *grad_outs):
        n = ctx.n_heads
        saved = ctx.saved_tensors
        x, hx, cx, body_w, body_b, w_ih, w_hh, b_g = saved[:8]
        head_ws = list(saved[8 : 8 + n])
        head_bs = list(saved[8 + n : 8 + 2 * n])
        stash = saved[8 + 2 * n]
        grad_heads = list(grad_outs[:n])
        grads = ext().seq_lstm_backward(
            x, hx, cx, body_w, body_b, w_ih, w_hh, b_g, head_ws, head_bs,
            stash, grad_heads,
        )
        # grads: dx, dhx, dcx, dbody_w, dbody_b, dw_ih, dw_hh, db_g,
        #        dhead_w..., dhead_b...
        return (None, *grads)


def seq_lstm_apply(core, x, hx, cx):
    params = (
        [core.body_w, core.body_b, core.w_ih, core.w_hh, core.b_g]
        + [p for n in core.head_names for p in core.head_params(n)]
    )
    res = _SeqLSTMFunction.apply(core, x, hx, cx, *params)
    n = len(core.head_names)
    outs = {name: res[i] for i, name in enumerate(core.head_names)}
    return outs, res[n], res[n + 1]
