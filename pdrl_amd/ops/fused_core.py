"""Autograd wiring for the fused SeqLSTMCore HIP kernels.

Forward: ONE kernel computes body GEMM+ReLU, the whole LSTM sequence, and all
heads (concatenated along the output dim), stashing activations for backward.
Backward: ONE kernel runs BPTT through heads/recurrence/body per batch row
(emitting per-step pre-activation gate grads), then the weight gradients are
plain library GEMMs (hipBLASLt via torch.matmul) over the stashed
activations — GEMM-shaped reductions belong on the matrix cores, and library
GEMMs are the sanctioned path for plain GEMMs.
"""
from __future__ import annotations

import torch

from . import ext


class _SeqLSTMFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, hx, cx, body_w, body_b, w_ih, w_hh, b_g, *head_params):
        head_ws = head_params[0::2]
        head_bs = head_params[1::2]
        heads_w = torch.cat(list(head_ws), dim=1).contiguous()
        heads_b = torch.cat(list(head_bs), dim=0).contiguous()
        x = x.contiguous()
        outs, hS, cS, stash = ext().seq_lstm_forward(
            x, hx.contiguous(), cx.contiguous(),
            body_w, body_b, w_ih, w_hh, b_g, heads_w, heads_b,
        )
        ctx.save_for_backward(x, hx, cx, body_w, w_ih, w_hh, heads_w, stash)
        ctx.head_dims = [int(w.shape[1]) for w in head_ws]
        return outs, hS, cS

    @staticmethod
    def backward(ctx, gouts, ghS, gcS):
        x, hx, cx, body_w, w_ih, w_hh, heads_w, stash = ctx.saved_tensors
        B, S, F = x.shape
        H = hx.shape[1]
        gouts = gouts.contiguous()
        ghS_c = ghS.contiguous() if ghS is not None else None
        gcS_c = gcS.contiguous() if gcS is not None else None

        dx, dh0, dc0, dgates, dxb = ext().seq_lstm_backward_core(
            gouts, ghS_c, gcS_c, stash, x, cx.contiguous(), body_w, w_ih, w_hh,
            heads_w,
        )

        # weight-gradient GEMMs over stashed activations (hipBLASLt)
        xb = stash[:, :, :H].reshape(-1, H)
        hseq = stash[:, :, 6 * H : 7 * H]
        hprev = torch.cat([hx.unsqueeze(1), hseq[:, :-1]], dim=1).reshape(-1, H)
        dg_flat = dgates.reshape(-1, 4 * H)
        dW_ih = xb.t() @ dg_flat
        dW_hh = hprev.t() @ dg_flat
        db_g = dg_flat.sum(0)

        dxb_flat = dxb.reshape(-1, H)
        dbody_w = x.reshape(-1, F).t() @ dxb_flat
        dbody_b = dxb_flat.sum(0)

        D = gouts.shape[-1]
        go_flat = gouts.reshape(-1, D)
        dheads_w = hseq.reshape(-1, H).t() @ go_flat
        dheads_b = go_flat.sum(0)

        head_grads = []
        off = 0
        for d in ctx.head_dims:
            head_grads.append(dheads_w[:, off : off + d].contiguous())
            head_grads.append(dheads_b[off : off + d].contiguous())
            off += d

        return (dx, dh0, dc0, dbody_w, dbody_b, dW_ih, dW_hh, db_g, *head_grads)


def seq_lstm_apply(core, x, hx, cx):
    head_params = [p for n in core.head_names for p in core.head_params(n)]
    outs_cat, hS, cS = _SeqLSTMFunction.apply(
        x, hx, cx, core.body_w, core.body_b, core.w_ih, core.w_hh, core.b_g,
        *head_params,
    )
    outs = {}
    off = 0
    for name in core.head_names:
        d = core.head_params(name)[0].shape[1]
        outs[name] = outs_cat[:, :, off : off + d]
        off += d
    return outs, hS, cS
