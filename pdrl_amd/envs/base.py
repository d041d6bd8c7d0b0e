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


def make(env_name: str, **kwargs):
    """gym.make-equivalent for the built-in environment set."""
    from . import cartpole, mountain_car, fake  # noqa: F401  (populate registry)

    if env_name not in _REGISTRY:
        raise ValueError(f"unknown env '{env_name}'; available: {sorted(_REGISTRY)}")
    return _REGISTRY[env_name](**kwargs)
