"""Flat-buffer fused optimizers (K13+K14 in SURVEY.md §2.4).

``FlatParamSpace`` re-homes a parameter group into ONE contiguous fp32 param
buffer and ONE contiguous grad buffer (parameters become views). Then:

* the whole optimizer epilogue is 2 kernels — l2norm² reduction + fused
  clip-and-update (RMSprop or Adam) — with the clip scale read from a
  device-resident scalar (no host sync, hipGraph-capturable);
* the flat grad buffer IS the single RCCL all-reduce bucket for the
  data-parallel learner (no gather/scatter copies at all).

Numerics match torch.optim.RMSprop/Adam + clip_grad_norm_ (GPU parity tests
in tests/test_gpu_kernels.py).
"""
from __future__ import annotations

import torch

from . import ext


class FlatParamSpace:
    def __init__(self, params):
        self.params = [p for p in params if p.requires_grad]
        assert self.params, "empty parameter group"
        device = self.params[0].device
        dtype = self.params[0].dtype
        assert dtype == torch.float32
        self.numel = sum(p.numel() for p in self.params)
        self.flat_param = torch.empty(self.numel, dtype=dtype, device=device)
        self.flat_grad = torch.zeros(self.numel, dtype=dtype, device=device)
        off = 0
        for p in self.params:
            n = p.numel()
            self.flat_param[off : off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off : off + n].view_as(p.data)
            p.grad = self.flat_grad[off : off + n].view_as(p.data)
            off += n


class _FusedOptimizer:
    """Common flat-space machinery; subclasses implement _update()."""

    is_fused = True

    def __init__(self, params, lr: float, max_norm: float | None):
        self.space = FlatParamSpace(params)
        self.lr = float(lr)
        self.max_norm = float(max_norm) if max_norm is not None else -1.0
        dev = self.space.flat_param.device
        self.norm_sq = torch.zeros(1, dtype=torch.float32, device=dev)

    @property
    def flat_grad(self):
        return self.space.flat_grad

    @property
    def flat_param(self):
        return self.space.flat_param

    def zero_grad(self, set_to_none: bool = False):
        # views must stay alive: always zero in place, never drop to None
        self.space.flat_grad.zero_()

    def grad_norm(self) -> torch.Tensor:
        """Device-resident L2 norm of the flat gradient (no host sync)."""
        self.norm_sq.zero_()
        ext().l2norm_sq(self.space.flat_grad, self.norm_sq)
        return self.norm_sq.sqrt()

    def step(self):
        self.norm_sq.zero_()
        ext().l2norm_sq(self.space.flat_grad, self.norm_sq)
        self._update()

    def state_dict(self):
        return {k: v for k, v in self._state().items()}

    def load_state_dict(self, sd):
        for k, v in self._state().items():
            if k in sd:
                v.copy_(sd[k].to(v.device))

    def _state(self) -> dict:
        raise NotImplementedError

    def _update(self):
        raise NotImplementedError


class FusedRMSprop(_FusedOptimizer):
    """torch.optim.RMSprop(lr, alpha=0.99, eps) + clip_grad_norm_ fused."""

    def __init__(self, params, lr, alpha=0.99, eps=1e-5, max_norm=None):
        super().__init__(params, lr, max_norm)
        self.alpha = float(alpha)
        self.eps = float(eps)
        self.sq_avg = torch.zeros_like(self.space.flat_param)

    def _state(self):
        return {"sq_avg": self.sq_avg}

    def _update(self):
        ext().rmsprop_step(
            self.space.flat_param, self.space.flat_grad, self.sq_avg,
            self.norm_sq, self.lr, self.alpha, self.eps, self.max_norm,
        )


class FusedAdam(_FusedOptimizer):
    """torch.optim.Adam(lr, betas, eps) + clip_grad_norm_ fused.
    Step count + bias corrections live on device (graph-replay safe).

    ``clock``: an optional SHARED state3 tensor (t, bc1, bc2). Optimizers
    that always step together once per iteration can share one clock;
    exactly one of them is the ``clock_owner`` and advances it — inside its
    own ``step()``, so the EAGER path stays correct as long as the owner
    steps first each iteration (the updaters' fixed actor→alpha→critic
    order guarantees that). Fused DAGs tick via AdamMultiGroup/step()."""

    def __init__(self, params, lr, betas=(0.9, 0.999), eps=1e-8, max_norm=None,
                 clock: torch.Tensor | None = None, clock_owner: bool = False):
        super().__init__(params, lr, max_norm)
        self.beta1, self.beta2 = float(betas[0]), float(betas[1])
        self.eps = float(eps)
        self.exp_avg = torch.zeros_like(self.space.flat_param)
        self.exp_avg_sq = torch.zeros_like(self.space.flat_param)
        self.shared_clock = clock is not None
        self.clock_owner = clock_owner or clock is None
        self.state3 = clock if clock is not None else torch.zeros(
            3, dtype=torch.float32, device=self.space.flat_param.device)

    def _state(self):
        return {"exp_avg": self.exp_avg, "exp_avg_sq": self.exp_avg_sq,
                "state3": self.state3}

    def tick(self):
        """Advance the device step clock (once per iteration when shared)."""
        ext().adam_prep(self.state3, self.beta1, self.beta2)

    def step(self):
        if self.shared_clock and self.clock_owner:
            self.tick()
        super().step()

    def _update(self, prep: bool | None = None):
        if prep is None:
            # shared clock: the owner ticks it (step()/AdamMultiGroup);
            # private clock: prep inline as before
            prep = not self.shared_clock
        ext().adam_step(
            self.space.flat_param, self.space.flat_grad, self.exp_avg,
            self.exp_avg_sq, self.state3, self.norm_sq, self.lr, self.beta1,
            self.beta2, self.eps, self.max_norm,
            do_prep=prep,
        )


class AdamMultiGroup:
    """ONE launch updating several FusedAdam groups that share a step clock
    (pointer + config tables prebuilt on device; graph-capture safe)."""

    def __init__(self, opts: list[FusedAdam]):
        assert opts and all(o.shared_clock for o in opts)
        assert all(o.state3 is opts[0].state3 for o in opts)
        assert all((o.beta1, o.beta2, o.eps) ==
                   (opts[0].beta1, opts[0].beta2, opts[0].eps) for o in opts)
        self.opts = opts
        self.state3 = opts[0].state3
        self.beta1, self.beta2, self.eps = (
            opts[0].beta1, opts[0].beta2, opts[0].eps)
        dev = opts[0].flat_param.device
        rows = []
        cfg = []
        for o in opts:
            nsq = o.norm_sq.data_ptr() if o.max_norm > 0 else 0
            rows.append([o.flat_param.data_ptr(), o.flat_grad.data_ptr(),
                         o.exp_avg.data_ptr(), o.exp_avg_sq.data_ptr(), nsq])
            cfg.append([float(o.space.numel), o.lr, o.max_norm])
        self.ptrs = torch.tensor(rows, dtype=torch.int64).to(dev)
        self.cfg = torch.tensor(cfg, dtype=torch.float32).to(dev)
        self.max_numel = max(o.space.numel for o in opts)
        self.owner = next((o for o in opts if o.clock_owner), opts[0])

    def update(self, tick: bool = True):
        """Apply all member updates; ``tick=False`` when an earlier kernel
        in the DAG already advanced the shared clock (the loss kernels can
        fold the prep in — saves the 1-thread launch)."""
        if tick:
            self.owner.tick()
        ext().adam_multi(self.ptrs, self.cfg, self.state3, len(self.opts),
                         self.max_numel, self.beta1, self.beta2, self.eps)
