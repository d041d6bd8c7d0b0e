"""MountainCarContinuous-v0: native implementation (standard published
dynamics per the Gymnasium MountainCarContinuous docs). Solved ≈ +90 reward.
"""
from __future__ import annotations

import math

import numpy as np

from .base import Box, register


@register("MountainCarContinuous-v0")
class MountainCarContinuousEnv:
    MIN_POSITION = -1.2
    MAX_POSITION = 0.6
    MAX_SPEED = 0.07
    GOAL_POSITION = 0.45
    GOAL_VELOCITY = 0.0
    POWER = 0.0015
    MAX_EPISODE_STEPS = 999

    def __init__(self, seed: int | None = None):
        self.observation_space = Box(
            np.array([self.MIN_POSITION, -self.MAX_SPEED], dtype=np.float32),
            np.array([self.MAX_POSITION, self.MAX_SPEED], dtype=np.float32),
        )
        self.action_space = Box(np.array([-1.0]), np.array([1.0]))
        self._rng = np.random.default_rng(seed)
        self._state = None
        self._steps = 0

    def seed(self, seed: int):
        self._rng = np.random.default_rng(seed)

    def reset(self, seed: int | None = None):
        if seed is not None:
            self.seed(seed)
        pos = self._rng.uniform(-0.6, -0.4)
        self._state = np.array([pos, 0.0], dtype=np.float64)
        self._steps = 0
        return self._state.astype(np.float32).copy(), {}

    def step(self, action):
        action = np.asarray(action, dtype=np.float64).reshape(-1)
        force = float(np.clip(action[0], -1.0, 1.0))
        position, velocity = self._state

        velocity += force * self.POWER - 0.0025 * math.cos(3 * position)
        velocity = float(np.clip(velocity, -self.MAX_SPEED, self.MAX_SPEED))
        position += velocity
        position = float(np.clip(position, self.MIN_POSITION, self.MAX_POSITION))
        if position <= self.MIN_POSITION and velocity < 0:
            velocity = 0.0
        self._state = np.array([position, velocity], dtype=np.float64)
        self._steps += 1

        terminated = bool(position >= self.GOAL_POSITION and velocity >= self.GOAL_VELOCITY)
        truncated = self._steps >= self.MAX_EPISODE_STEPS
        reward = -0.1 * force**2
        if terminated:
            reward += 100.0
        return self._state.astype(np.float32).copy(), reward, terminated, truncated, {}
