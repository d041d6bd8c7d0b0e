"""FakeEnv: deterministic synthetic transition generator for pipeline tests
(the test strategy SURVEY.md §4 calls for — validates the actor/transport/
assembler path without real env dynamics).
"""
from __future__ import annotations

import numpy as np

from .base import Box, Discrete, register


@register("Fake-v0")
class FakeEnv:
    """Fixed-seed synthetic env: obs is a deterministic function of the step
    counter, episodes end after ``episode_len`` steps."""

    def __init__(self, obs_dim: int = 4, n_actions: int = 2, episode_len: int = 17, seed: int = 0):
        self.observation_space = Box(
            -np.ones(obs_dim, dtype=np.float32) * 10, np.ones(obs_dim, dtype=np.float32) * 10
        )
        self.action_space = Discrete(n_actions)
        self.episode_len = episode_len
        self.MAX_EPISODE_STEPS = episode_len
        self._seed = seed
        self._t = 0
        self._episode = 0

    def seed(self, seed: int):
        self._seed = seed

    def _obs(self):
        base = np.arange(self.observation_space.shape[0], dtype=np.float32)
        return np.sin(base + 0.1 * self._t + self._episode + self._seed).astype(np.float32)

    def reset(self, seed: int | None = None):
        if seed is not None:
            self.seed(seed)
        self._t = 0
        self._episode += 1
        return self._obs(), {}

    def step(self, action):
        self._t += 1
        terminated = False
        truncated = self._t >= self.episode_len
        reward = 1.0
        return self._obs(), reward, terminated, truncated, {}
