"""Shared updater machinery: gradient reduce + clip, initial recurrent state,
checkpoint helpers.
"""
from __future__ import annotations

import torch


def batch_initial_state(batch: dict[str, torch.Tensor]):
    """The batch's initial LSTM state: stored per-step hx/cx at t=0
    (reference: each learning.py uses hx[:, 0], cx[:, 0] — e.g.
    ppo/learning.py:37-41)."""
    return batch["hx"][:, 0].contiguous(), batch["cx"][:, 0].contiguous()


class BaseUpdater:
    """Common skeleton: owns hyperparams, optional data-parallel gradient
    reducer, grad-norm clipping, and an update counter."""

    name = "base"

    def __init__(self, params, device, grad_reducer=None):
        self.params = params
        self.device = torch.device(device)
        self.grad_reducer = grad_reducer
        self.update_count = 0

    # -- to be provided by subclasses ----------------------------------- #
    def trainable_modules(self) -> dict[str, torch.nn.Module]:
        raise NotImplementedError

    def optimizers(self) -> dict[str, torch.optim.Optimizer]:
        raise NotImplementedError

    def step(self, batch) -> dict:
        raise NotImplementedError

    # -- shared --------------------------------------------------------- #
    def make_optimizer(self, kind: str, parameters, lr: float,
                       clip: bool = True, **kw):
        """Fused flat-buffer optimizer (HIP, clip folded in) on GPU with the
        extension loaded; stock torch.optim elsewhere."""
        parameters = list(parameters)
        from pdrl_amd import ops

        max_norm = self.params.max_grad_norm if clip else None
        if self.device.type == "cuda" and ops.available():
            from pdrl_amd.ops.optim import FusedAdam, FusedRMSprop

            if kind == "rmsprop":
                return FusedRMSprop(parameters, lr=lr, max_norm=max_norm,
                                    eps=kw.get("eps", 1e-5))
            return FusedAdam(parameters, lr=lr, max_norm=max_norm,
                             clock=kw.get("clock"),
                             clock_owner=kw.get("clock_owner", False))
        if kind == "rmsprop":
            return torch.optim.RMSprop(parameters, lr=lr, eps=kw.get("eps", 1e-5))
        return torch.optim.Adam(parameters, lr=lr)

    def make_fused_step(self, algo: str, model, optimizer, duals=None):
        """Whole-step fused HIP DAG (IMPALA/PPO, discrete policies) when the
        extension is loaded on GPU; None → the eager-autograd path runs."""
        from pdrl_amd import ops

        if not (self.device.type == "cuda" and ops.available()):
            return None
        if not getattr(optimizer, "is_fused", False):
            return None
        actor = getattr(model, "actor", model)
        core = getattr(actor, "core", None)
        if core is None:
            return None
        if core.head_names == ["mu", "std", "value"] and algo == "PPO":
            algo = "PPO-C"  # Gaussian-policy fused loss (K5)
        elif core.head_names != ["logits", "value"]:
            return None
        from pdrl_amd.ops.fused_step import FusedOnPolicyStep

        # Graph policy: single-rank captures the whole step. Multi-rank
        # (decided inside FusedOnPolicyStep.run) defaults to SPLIT-GRAPH —
        # two captured graphs around a stream-ordered RCCL all-reduce —
        # because full capture including the collective cannot be validated
        # before the driver's one multi-GPU run; PDRL_GRAPH_RCCL=1 opts
        # into full capture, PDRL_USE_GRAPH=0 forces stream-ordered.
        import os

        use_graph = bool(int(os.environ.get("PDRL_USE_GRAPH", "1")))
        return FusedOnPolicyStep(algo, core, self.params, optimizer,
                                 grad_reducer=self.grad_reducer,
                                 use_graph=use_graph, duals=duals)

    def apply_step(self, optimizer, parameters):
        """Gradient epilogue: all-reduce across ranks → clip → update.
        Fused path: one collective on the flat grad bucket; clip is inside
        the update kernel. Eager path: per-grad reduce + torch clip."""
        if getattr(optimizer, "is_fused", False):
            if self.grad_reducer is not None:
                self.grad_reducer.all_reduce([optimizer.flat_grad])
            optimizer.step()
        else:
            self.reduce_and_clip(parameters)
            optimizer.step()

    def reduce_and_clip(self, parameters):
        """All-reduce gradients across learner ranks (RCCL on GPU, gloo on
        CPU), then global-norm clip. Called between backward() and step()."""
        parameters = [p for p in parameters if p.grad is not None]
        if self.grad_reducer is not None:
            self.grad_reducer.all_reduce([p.grad for p in parameters])
        torch.nn.utils.clip_grad_norm_(parameters, self.params.max_grad_norm)

    def actor_state_dict(self):
        """State broadcast to workers (the policy network only)."""
        mods = self.trainable_modules()
        m = mods.get("model") or next(iter(mods.values()))
        actor = getattr(m, "actor", m)
        return {k: v.cpu() for k, v in actor.state_dict().items()}

    def extra_params(self) -> dict:
        """Loose nn.Parameters outside any module (SAC's log_alpha, V-MPO's
        duals): they must be checkpointed explicitly or they silently reset
        on resume."""
        return {}

    def save(self, path):
        torch.save(
            {
                "algo": self.name,
                "update_count": self.update_count,
                "modules": {k: m.state_dict() for k, m in self.trainable_modules().items()},
                "optimizers": {k: o.state_dict() for k, o in self.optimizers().items()},
                "extras": {k: p.detach().cpu() for k, p in self.extra_params().items()},
            },
            path,
        )

    def load(self, path, map_location="cpu"):
        ckpt = torch.load(path, map_location=map_location, weights_only=False)
        for k, m in self.trainable_modules().items():
            if k in ckpt.get("modules", {}):
                m.load_state_dict(ckpt["modules"][k])
        for k, p in self.extra_params().items():
            if k in ckpt.get("extras", {}):
                with torch.no_grad():
                    p.data.copy_(ckpt["extras"][k].to(p.device))
        for k, o in self.optimizers().items():
            if k in ckpt.get("optimizers", {}):
                try:
                    o.load_state_dict(ckpt["optimizers"][k])
                except Exception as e:  # e.g. fused↔eager optimizer switch
                    import warnings

                    warnings.warn(
                        f"checkpoint optimizer state for '{k}' not restored "
                        f"({type(e).__name__}: {e}); continuing with fresh "
                        "optimizer state")
        self.update_count = int(ckpt.get("update_count", 0))
        return ckpt
