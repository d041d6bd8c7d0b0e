"""Model zoo: MLP + LSTM actor/critic families for discrete and continuous
control.

Capability parity with the reference's networks/models.py (13 classes,
reference: networks/models.py:8-378) with an MI355X-first internal design:

* All variants are built on one ``SeqLSTMCore``: body Linear+ReLU → LSTM over
  the sequence → per-head Linears. Gate math is written out explicitly
  (sigmoid/tanh on fused gate GEMMs) rather than via nn.LSTMCell, and weights
  are stored TRANSPOSED — ``(in_features, out_features)`` — which is the
  layout the fused CDNA4 HIP kernel consumes (lane = output column, columns
  contiguous: coalesced global loads, conflict-free LDS). The eager path and
  the HIP path share one parameter layout, so the eager path is the numerics
  oracle for kernel parity tests.
* On GPU with the HIP extension loaded, ``SeqLSTMCore.forward`` dispatches to
  the fused kernel (one launch for body+LSTM seq+heads instead of the
  reference's per-step Python loop over LSTMCell — reference
  models.py:71-75 and its 4 clones). On CPU it runs the eager path.
* The SAC "separate" wrappers deliver genuinely independent critic/target
  parameters (the reference aliases critic and target-critic —
  learner.py:357 — a latent bug this framework does not replicate).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

LOG_STD_MIN, LOG_STD_MAX = -20.0, 2.0


def _init_linear_t(w: torch.Tensor, b: torch.Tensor | None, fan_in: int):
    bound = 1.0 / math.sqrt(fan_in) if fan_in > 0 else 0.0
    with torch.no_grad():
        w.uniform_(-bound, bound)
        if b is not None:
            b.uniform_(-bound, bound)


class SeqLSTMCore(nn.Module):
    """body Linear+ReLU → single-layer LSTM over seq dim → head Linears.

    Parameters (all transposed layout):
      body_w (F, H), body_b (H)
      w_ih (H, 4H), w_hh (H, 4H), b_g (4H)   # gate order i, f, g, o
      per head k: head_w[k] (H, D_k), head_b[k] (D_k)

    Dual-body mode (``input2_dim``): the LSTM input is the concatenation
    [relu(x·body_w+body_b) | relu(x2·body2_w+body2_b)], each encoder to
    H/2 — the reference's continuous-critic topology (obs encoder ‖ action
    encoder straight into the LSTM, reference: networks/models.py:273-322).
    Then body_w is (F, H/2) and body2_w (F2, H/2).
    """

    def __init__(self, input_dim: int, hidden: int, heads: dict[str, int],
                 input2_dim: int | None = None):
        super().__init__()
        self.input_dim = input_dim
        self.input2_dim = input2_dim
        self.hidden = hidden
        self.head_names = list(heads.keys())
        self.head_dims = dict(heads)
        H = hidden
        if input2_dim is None:
            self.body_w = nn.Parameter(torch.empty(input_dim, H))
            self.body_b = nn.Parameter(torch.empty(H))
            self.body2_w = None
            self.body2_b = None
        else:
            assert H % 2 == 0, "dual body needs an even hidden size"
            half = H // 2
            self.body_w = nn.Parameter(torch.empty(input_dim, half))
            self.body_b = nn.Parameter(torch.empty(half))
            self.body2_w = nn.Parameter(torch.empty(input2_dim, half))
            self.body2_b = nn.Parameter(torch.empty(half))
            _init_linear_t(self.body2_w, self.body2_b, input2_dim)
        self.w_ih = nn.Parameter(torch.empty(H, 4 * H))
        self.w_hh = nn.Parameter(torch.empty(H, 4 * H))
        self.b_g = nn.Parameter(torch.empty(4 * H))
        _init_linear_t(self.body_w, self.body_b, input_dim)
        _init_linear_t(self.w_ih, None, H)
        _init_linear_t(self.w_hh, self.b_g, H)
        # heads live CONCATENATED along the output dim: one (H, Dtot) GEMM in
        # both the eager and the fused path, and the parameter layout matches
        # the HIP weight-grad kernel exactly (no cats/splits anywhere).
        self.out_dim = sum(heads.values())
        self.heads_w = nn.Parameter(torch.empty(H, self.out_dim))
        self.heads_b = nn.Parameter(torch.empty(self.out_dim))
        _init_linear_t(self.heads_w, self.heads_b, H)
        self._head_slices = {}
        off = 0
        for name, dim in heads.items():
            self._head_slices[name] = (off, off + dim)
            off += dim

    def head_params(self, name: str):
        """(weight, bias) views of the concatenated head parameters."""
        lo, hi = self._head_slices[name]
        return self.heads_w[:, lo:hi], self.heads_b[lo:hi]

    def split_heads(self, outs_cat: torch.Tensor) -> dict[str, torch.Tensor]:
        return {
            name: outs_cat[..., lo:hi] for name, (lo, hi) in self._head_slices.items()
        }

    # ------------------------------------------------------------------ #
    def forward(
        self, x: torch.Tensor, hx: torch.Tensor, cx: torch.Tensor,
        x2: torch.Tensor | None = None,
    ) -> tuple[dict[str, torch.Tensor], torch.Tensor, torch.Tensor]:
        """x: (B, S, F); hx/cx: (B, H). Returns ({head: (B,S,D)}, h_S, c_S).
        Dual-body cores additionally take x2: (B, S, F2)."""
        assert (x2 is None) == (self.input2_dim is None), "dual-body mismatch"
        if x.is_cuda:
            from pdrl_amd import ops

            # the MFMA wgrad kernels stage a (B*S)-entry row-pointer table
            # in LDS (64 KB) — beyond that, degrade to the eager torch path
            # instead of raising mid-training (profiles/batch_scaling_r02.md)
            if ops.available() and x.shape[0] * x.shape[1] <= 8192:
                return self._forward_fused(x, hx, cx, x2)
        return self._forward_eager(x, hx, cx, x2)

    def _forward_eager(self, x, hx, cx, x2=None):
        B, S, Fdim = x.shape
        H = self.hidden
        if x2 is None:
            xb = F.relu(x.reshape(B * S, Fdim) @ self.body_w + self.body_b)
            xb = xb.view(B, S, H)
        else:
            o = F.relu(x.reshape(B * S, Fdim) @ self.body_w + self.body_b)
            a = F.relu(x2.reshape(B * S, -1) @ self.body2_w + self.body2_b)
            xb = torch.cat([o, a], dim=-1).view(B, S, H)
        h, c = hx, cx
        hs = []
        for t in range(S):
            gates = xb[:, t] @ self.w_ih + h @ self.w_hh + self.b_g
            gi, gf, gg, go = gates.chunk(4, dim=1)
            gi, gf, go = torch.sigmoid(gi), torch.sigmoid(gf), torch.sigmoid(go)
            gg = torch.tanh(gg)
            c = gf * c + gi * gg
            h = go * torch.tanh(c)
            hs.append(h)
        hseq = torch.stack(hs, dim=1)  # (B, S, H)
        outs_cat = (hseq.reshape(B * S, H) @ self.heads_w + self.heads_b).view(
            B, S, self.out_dim
        )
        return self.split_heads(outs_cat), h, c

    def _forward_fused(self, x, hx, cx, x2=None):
        from pdrl_amd import ops

        return ops.seq_lstm_forward(self, x, hx, cx, x2)

    @torch.no_grad()
    def step(self, x: torch.Tensor, hx: torch.Tensor, cx: torch.Tensor,
             x2: torch.Tensor | None = None):
        """Single-step inference (actor-side). x: (B, F) → ({head:(B,D)}, h, c)."""
        outs, h, c = self._forward_eager(
            x.unsqueeze(1), hx, cx, x2.unsqueeze(1) if x2 is not None else None)
        return {k: v.squeeze(1) for k, v in outs.items()}, h, c


# --------------------------------------------------------------------------- #
# Distribution helpers (eager oracle; fused into HIP loss kernels on GPU)
# --------------------------------------------------------------------------- #
def categorical_stats(logits: torch.Tensor, actions: torch.Tensor):
    """log π(a), entropy, log-softmax for discrete policies.

    logits: (..., A); actions: (..., 1) int64 → (log_prob (...,1), entropy (...,1))
    """
    logp = F.log_softmax(logits, dim=-1)
    p = logp.exp()
    entropy = -(p * logp).sum(-1, keepdim=True)
    log_prob = logp.gather(-1, actions.long())
    return log_prob, entropy


def sample_categorical(logits: torch.Tensor, generator=None):
    probs = F.softmax(logits, dim=-1)
    return torch.multinomial(probs, 1, generator=generator)


# --------------------------------------------------------------------------- #
# Shared-torso actor-critic (PPO / IMPALA / V-MPO)
# --------------------------------------------------------------------------- #
class MlpLSTMBase(nn.Module):
    """Discrete shared actor-critic (reference: networks/models.py:8-100)."""

    def __init__(self, f: int, n_outputs: int, seq_len: int, hidden_size: int):
        super().__init__()
        self.input_dim = f
        self.n_outputs = n_outputs
        self.seq_len = seq_len
        self.hidden_size = hidden_size
        self.core = SeqLSTMCore(f, hidden_size, {"logits": n_outputs, "value": 1})

    @torch.no_grad()
    def act(self, obs: torch.Tensor, lstm_hxs: tuple[torch.Tensor, torch.Tensor]):
        """Single env step: obs (1, F) → (action (1,1), logits (1,A),
        log_prob (1,1), (hx, cx))."""
        outs, h, c = self.core.step(obs, *lstm_hxs)
        logits = outs["logits"]
        action = sample_categorical(logits)
        log_prob, _ = categorical_stats(logits, action)
        return action, logits, log_prob, (h.detach(), c.detach())

    def forward(self, obs: torch.Tensor, lstm_hxs, behaviour_acts: torch.Tensor):
        """Batched training forward: obs (B,S,F), behaviour_acts (B,S,1) →
        (logits (B,S,A), log_probs (B,S,1), entropy (B,S,1), value (B,S,1))."""
        outs, _, _ = self.core(obs, *lstm_hxs)
        logits, value = outs["logits"], outs["value"]
        log_probs, entropy = categorical_stats(logits, behaviour_acts)
        return logits, log_probs, entropy, value


class MlpLSTMContinuous(nn.Module):
    """Continuous shared actor-critic: Normal(tanh(mu), softplus(std))
    (reference: networks/models.py:103-118)."""

    def __init__(self, f: int, n_outputs: int, seq_len: int, hidden_size: int):
        super().__init__()
        self.input_dim = f
        self.n_outputs = n_outputs
        self.seq_len = seq_len
        self.hidden_size = hidden_size
        self.core = SeqLSTMCore(
            f, hidden_size, {"mu": n_outputs, "std": n_outputs, "value": 1}
        )

    def _dist(self, mu, std):
        # validate_args=False: the validation reduction is a host sync, which
        # both costs latency and forbids hipGraph capture of the step
        return torch.distributions.Normal(
            torch.tanh(mu), F.softplus(std) + 1e-4, validate_args=False
        )

    @torch.no_grad()
    def act(self, obs, lstm_hxs):
        outs, h, c = self.core.step(obs, *lstm_hxs)
        dist = self._dist(outs["mu"], outs["std"])
        action = dist.sample()
        log_prob = dist.log_prob(action).sum(-1, keepdim=True)
        return action, torch.cat([outs["mu"], outs["std"]], -1), log_prob, (h, c)

    def forward(self, obs, lstm_hxs, behaviour_acts):
        outs, _, _ = self.core(obs, *lstm_hxs)
        dist = self._dist(outs["mu"], outs["std"])
        log_probs = dist.log_prob(behaviour_acts).sum(-1, keepdim=True)
        entropy = dist.entropy().sum(-1, keepdim=True)
        logits = torch.cat([outs["mu"], outs["std"]], -1)
        return logits, log_probs, entropy, outs["value"]


# --------------------------------------------------------------------------- #
# SAC actors
# --------------------------------------------------------------------------- #
class MlpLSTMActor(nn.Module):
    """SAC-discrete actor: π(a|s) probabilities
    (reference: networks/models.py:121-159)."""

    def __init__(self, f: int, n_outputs: int, seq_len: int, hidden_size: int):
        super().__init__()
        self.input_dim = f
        self.n_outputs = n_outputs
        self.seq_len = seq_len
        self.hidden_size = hidden_size
        self.core = SeqLSTMCore(f, hidden_size, {"logits": n_outputs})

    @torch.no_grad()
    def act(self, obs, lstm_hxs):
        outs, h, c = self.core.step(obs, *lstm_hxs)
        logits = outs["logits"]
        action = sample_categorical(logits)
        log_prob, _ = categorical_stats(logits, action)
        return action, logits, log_prob, (h, c)

    def forward(self, obs, lstm_hxs):
        """→ (probs (B,S,A), log_probs (B,S,A)) with zero-prob guard."""
        outs, _, _ = self.core(obs, *lstm_hxs)
        logits = outs["logits"]
        probs = F.softmax(logits, dim=-1)
        zero_mask = (probs == 0.0).float()
        log_probs = torch.log(probs + zero_mask * 1e-8)
        return probs, log_probs


class MlpLSTMActorContinuous(nn.Module):
    """SAC-continuous actor: tanh-squashed reparameterized Gaussian
    (reference: networks/models.py:162-231)."""

    def __init__(self, f: int, n_outputs: int, seq_len: int, hidden_size: int):
        super().__init__()
        self.input_dim = f
        self.n_outputs = n_outputs
        self.seq_len = seq_len
        self.hidden_size = hidden_size
        self.core = SeqLSTMCore(f, hidden_size, {"mu": n_outputs, "log_std": n_outputs})

    def _sample(self, mu, log_std, reparam: bool):
        log_std = torch.clamp(log_std, LOG_STD_MIN, LOG_STD_MAX)
        std = log_std.exp()
        dist = torch.distributions.Normal(mu, std, validate_args=False)
        z = dist.rsample() if reparam else dist.sample()
        action = torch.tanh(z)
        log_prob = dist.log_prob(z) - torch.log(1.0 - action.pow(2) + 1e-7)
        return action, log_prob.sum(-1, keepdim=True)

    @torch.no_grad()
    def act(self, obs, lstm_hxs):
        outs, h, c = self.core.step(obs, *lstm_hxs)
        action, log_prob = self._sample(outs["mu"], outs["log_std"], reparam=False)
        logits = torch.cat([outs["mu"], outs["log_std"]], -1)
        return action, logits, log_prob, (h, c)

    def forward(self, obs, lstm_hxs):
        """→ (action (B,S,A) with reparam grad, log_prob (B,S,1))."""
        outs, _, _ = self.core(obs, *lstm_hxs)
        return self._sample(outs["mu"], outs["log_std"], reparam=True)


# --------------------------------------------------------------------------- #
# SAC critics
# --------------------------------------------------------------------------- #
class MlpLSTMCritic(nn.Module):
    """SAC-discrete critic Q(s, ·) (reference: networks/models.py:234-270)."""

    def __init__(self, f: int, n_outputs: int, seq_len: int, hidden_size: int):
        super().__init__()
        self.input_dim = f
        self.n_outputs = n_outputs
        self.seq_len = seq_len
        self.hidden_size = hidden_size
        self.core = SeqLSTMCore(f, hidden_size, {"q": n_outputs})

    def forward(self, obs, lstm_hxs):
        outs, _, _ = self.core(obs, *lstm_hxs)
        return outs["q"]


class MlpLSTMCriticContinuous(nn.Module):
    """SAC-continuous critic Q(s, a): obs and action encoded to half-hidden
    each, concatenated straight into the LSTM — the reference topology
    (networks/models.py:273-322), realized as a dual-body SeqLSTMCore so
    the whole critic forward is ONE fused kernel launch on GPU."""

    def __init__(self, f: int, n_outputs: int, seq_len: int, hidden_size: int):
        super().__init__()
        assert hidden_size % 2 == 0
        self.input_dim = f
        self.n_outputs = n_outputs
        self.seq_len = seq_len
        self.hidden_size = hidden_size
        self.core = SeqLSTMCore(f, hidden_size, {"q": 1}, input2_dim=n_outputs)

    def forward(self, obs, act, lstm_hxs):
        outs, _, _ = self.core(obs, *lstm_hxs, x2=act)
        return outs["q"]


class MlpLSTMDoubleCritic(nn.Module):
    """Twin-Q wrapper, discrete (reference: networks/models.py:325-333)."""

    def __init__(self, *args, **kwargs):
        super().__init__()
        self.q1 = MlpLSTMCritic(*args, **kwargs)
        self.q2 = MlpLSTMCritic(*args, **kwargs)
        self.n_outputs = self.q1.n_outputs

    def forward(self, obs, lstm_hxs):
        return self.q1(obs, lstm_hxs), self.q2(obs, lstm_hxs)


class MlpLSTMDoubleCriticContinuous(nn.Module):
    """Twin-Q wrapper, continuous (reference: networks/models.py:336-342)."""

    def __init__(self, *args, **kwargs):
        super().__init__()
        self.q1 = MlpLSTMCriticContinuous(*args, **kwargs)
        self.q2 = MlpLSTMCriticContinuous(*args, **kwargs)
        self.n_outputs = self.q1.n_outputs

    def forward(self, obs, act, lstm_hxs):
        return self.q1(obs, act, lstm_hxs), self.q2(obs, act, lstm_hxs)


# --------------------------------------------------------------------------- #
# Top-level wrappers selected by the runner
# --------------------------------------------------------------------------- #
class MlpLSTMSingle(nn.Module):
    """PPO/IMPALA/V-MPO wrapper: one shared actor-critic; ``critic`` aliases
    ``actor`` by design (single-network algorithms)
    (reference: networks/models.py:345-352)."""

    def __init__(self, f, n_outputs, seq_len, hidden_size):
        super().__init__()
        self.actor = MlpLSTMBase(f, n_outputs, seq_len, hidden_size)
        self.critic = self.actor

    @torch.no_grad()
    def act(self, obs, lstm_hxs):
        return self.actor.act(obs, lstm_hxs)


class MlpLSTMSingleContinuous(nn.Module):
    """Continuous counterpart (reference: networks/models.py:355-360)."""

    def __init__(self, f, n_outputs, seq_len, hidden_size):
        super().__init__()
        self.actor = MlpLSTMContinuous(f, n_outputs, seq_len, hidden_size)
        self.critic = self.actor

    @torch.no_grad()
    def act(self, obs, lstm_hxs):
        return self.actor.act(obs, lstm_hxs)


class MlpLSTMSeperate(nn.Module):
    """SAC-discrete wrapper: independent actor + twin critic
    (reference: networks/models.py:363-369)."""

    def __init__(self, f, n_outputs, seq_len, hidden_size):
        super().__init__()
        self.actor = MlpLSTMActor(f, n_outputs, seq_len, hidden_size)
        self.critic = MlpLSTMDoubleCritic(f, n_outputs, seq_len, hidden_size)

    @torch.no_grad()
    def act(self, obs, lstm_hxs):
        return self.actor.act(obs, lstm_hxs)


class MlpLSTMSeperateContinuous(nn.Module):
    """SAC-continuous wrapper (reference: networks/models.py:372-378)."""

    def __init__(self, f, n_outputs, seq_len, hidden_size):
        super().__init__()
        self.actor = MlpLSTMActorContinuous(f, n_outputs, seq_len, hidden_size)
        self.critic = MlpLSTMDoubleCriticContinuous(f, n_outputs, seq_len, hidden_size)

    @torch.no_grad()
    def act(self, obs, lstm_hxs):
        return self.actor.act(obs, lstm_hxs)
