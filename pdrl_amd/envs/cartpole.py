"""CartPole-v1: native implementation of the classic cart-pole balancing task
(standard published dynamics — Barto, Sutton & Anderson 1983, as specified by
the Gymnasium CartPole-v1 docs). Solved threshold: mean reward 475/500.
"""
from __future__ import annotations

import math

import numpy as np

from .base import Box, Discrete, register


@register("CartPole-v1")
class CartPoleEnv:
    GRAVITY = 9.8
    MASS_CART = 1.0
    MASS_POLE = 0.1
    TOTAL_MASS = MASS_CART + MASS_POLE
    LENGTH = 0.5  # half pole length
    POLE_MASS_LENGTH = MASS_POLE * LENGTH
    FORCE_MAG = 10.0
    TAU = 0.02
    THETA_THRESHOLD = 12 * 2 * math.pi / 360
    X_THRESHOLD = 2.4
    MAX_EPISODE_STEPS = 500

    def __init__(self, seed: int | None = None):
        high = np.array(
            [
                self.X_THRESHOLD * 2,
                np.inf,
                self.THETA_THRESHOLD * 2,
                np.inf,
            ],
            dtype=np.float32,
        )
        self.observation_space = Box(-high, high)
        self.action_space = Discrete(2)
        self._rng = np.random.default_rng(seed)
        self._state = None
        self._steps = 0

    def seed(self, seed: int):
        self._rng = np.random.default_rng(seed)

    def reset(self, seed: int | None = None):
        if seed is not None:
            self.seed(seed)
        self._state = self._rng.uniform(-0.05, 0.05, size=4).astype(np.float64)
        self._steps = 0
        return self._state.astype(np.float32).copy(), {}

    def step(self, action):
        action = int(action)
        assert action in (0, 1), f"invalid action {action}"
        x, x_dot, theta, theta_dot = self._state
        force = self.FORCE_MAG if action == 1 else -self.FORCE_MAG
        cos_t = math.cos(theta)
        sin_t = math.sin(theta)

        temp = (force + self.POLE_MASS_LENGTH * theta_dot**2 * sin_t) / self.TOTAL_MASS
        theta_acc = (self.GRAVITY * sin_t - cos_t * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASS_POLE * cos_t**2 / self.TOTAL_MASS)
        )
        x_acc = temp - self.POLE_MASS_LENGTH * theta_acc * cos_t / self.TOTAL_MASS

        # Euler integration
        x = x + self.TAU * x_dot
        x_dot = x_dot + self.TAU * x_acc
        theta = theta + self.TAU * theta_dot
        theta_dot = theta_dot + self.TAU * theta_acc
        self._state = np.array([x, x_dot, theta, theta_dot], dtype=np.float64)
        self._steps += 1

        terminated = bool(
            x < -self.X_THRESHOLD
            or x > self.X_THRESHOLD
            or theta < -self.THETA_THRESHOLD
            or theta > self.THETA_THRESHOLD
        )
        truncated = self._steps >= self.MAX_EPISODE_STEPS
        reward = 1.0
        return self._state.astype(np.float32).copy(), reward, terminated, truncated, {}
