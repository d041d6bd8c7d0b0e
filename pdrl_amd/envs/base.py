"""Minimal gymnasium-compatible environment API.

The deployment container has no gymnasium, so the framework ships its own
implementations of the two environments the reference targets
(reference: README.md:19-21, utils/parameters.json "env") with the standard
``reset() -> (obs, info)`` / ``step(a) -> (obs, reward, terminated, truncated,
info)`` contract and ``observation_space`` / ``action_space`` members.
"""
from __future__ import annotations

import numpy as np


class Space:
    pass


class Discrete(Space):
    def __init__(self, n: int):
        self.n = int(n)
        self.shape = ()
        self.dtype = np.int64

    def sample(self, rng=None):
        rng = rng or np.random
        return int(rng.integers(self.n)) if hasattr(rng, "integers") else int(rng.randint(self.n))

    def __repr__(self):
        return f"Discrete({self.n})"


class Box(Space):
    def __init__(self, low, high, shape=None, dtype=np.float32):
        self.low = np.asarray(low, dtype=dtype)
        self.high = np.asarray(high, dtype=dtype)
        if shape is None:
            shape = np.broadcast(self.low, self.high).shape
        self.shape = tuple(shape)
        self.low = np.broadcast_to(self.low, self.shape).astype(dtype)
        self.high = np.broadcast_to(self.high, self.shape).astype(dtype)
        self.dtype = dtype

    def sample(self, rng=None):
        rng = rng or np.random
        low = np.where(np.isfinite(self.low), self.low, -1.0)
        high = np.where(np.isfinite(self.high), self.high, 1.0)
        u = rng.random(self.shape) if hasattr(rng, "random") else rng.rand(*self.shape)
        return (low + u * (high - low)).astype(self.dtype)

    def __repr__(self):
        return f"Box{self.shape}"


_REGISTRY = {}


def register(name):
    def deco(cls):
        _REGISTRY[name] = cls
        return cls

    return deco


def _convert_space(sp) -> Space:
    """Translate a gymnasium space into the native Space types the rest of
    the framework type-checks against (EnvBase / probe_env_spaces use
    ``isinstance(..., Box)``)."""
    if hasattr(sp, "n"):
        return Discrete(int(sp.n))
    if hasattr(sp, "low") and hasattr(sp, "high"):
        return Box(sp.low, sp.high, shape=getattr(sp, "shape", None))
    raise TypeError(f"unsupported gymnasium space type: {type(sp).__name__}")


class GymnasiumAdapter:
    """Any-gymnasium-env support behind the native registry (the reference
    wraps arbitrary gym envs — agents/worker_module/env_maker.py:6-31).
    Exposes the native API: ``seed()`` (applied at the next reset, the
    gymnasium convention), native Space objects, and the standard
    reset/step tuples, which gymnasium already produces."""

    def __init__(self, env):
        self.env = env
        self._pending_seed: int | None = None
        self.observation_space = _convert_space(env.observation_space)
        self.action_space = _convert_space(env.action_space)

    def seed(self, seed: int):
        self._pending_seed = int(seed)

    def reset(self):
        if self._pending_seed is not None:
            seed, self._pending_seed = self._pending_seed, None
            return self.env.reset(seed=seed)
        return self.env.reset()

    def step(self, action):
        return self.env.step(action)

    def close(self):
        if hasattr(self.env, "close"):
            self.env.close()


def make(env_name: str, **kwargs):
    """gym.make-equivalent: native envs first, then any gymnasium env when
    gymnasium is importable (absent in this container — the native
    CartPole/MountainCarContinuous implementations cover the reference's
    target envs without it)."""
    from . import cartpole, mountain_car, fake  # noqa: F401  (populate registry)

    if env_name in _REGISTRY:
        return _REGISTRY[env_name](**kwargs)
    try:
        import gymnasium
    except ImportError:
        raise ValueError(
            f"unknown env '{env_name}'; native envs: {sorted(_REGISTRY)} "
            "(gymnasium not importable, so arbitrary gym envs are unavailable)"
        ) from None
    return GymnasiumAdapter(gymnasium.make(env_name, **kwargs))
