"""EnvBase: environment adapter for the actor loop.

Capability parity with the reference's agents/worker_module/env_maker.py
(reset→preprocessed obs; step merges terminated|truncated into one done:
30-31; continuous Box actions unwrapped from tensors: 15-26) — built on the
framework's native env registry instead of gym.
"""
from __future__ import annotations

import numpy as np
import torch

from pdrl_amd import envs
from pdrl_amd.envs.base import Box
from pdrl_amd.utils import obs_preprocess


class EnvBase:
    def __init__(self, env_name: str, seed: int | None = None):
        self.env = envs.make(env_name)
        if seed is not None:
            self.env.seed(seed)
        self.observation_space = self.env.observation_space
        self.action_space = self.env.action_space
        self.continuous = isinstance(self.action_space, Box)

    def reset(self) -> torch.Tensor:
        obs, _ = self.env.reset()
        return obs_preprocess(obs)

    def _prepare_action(self, action):
        # fast paths for callers that already converted (the vectorized
        # worker passes plain ints / float32 rows — per-env tensor→numpy
        # conversion cost ~10% of the rollout tick in profiling)
        if type(action) is int and not self.continuous:
            return action
        if isinstance(action, np.ndarray) and self.continuous and \
                action.dtype == np.float32:
            return action.reshape(-1)
        if isinstance(action, torch.Tensor):
            action = action.detach().cpu().numpy()
        if self.continuous:
            return np.asarray(action, dtype=np.float32).reshape(-1)
        return int(np.asarray(action).reshape(-1)[0])

    def step(self, action):
        obs, rew, terminated, truncated, info = self.env.step(self._prepare_action(action))
        done = bool(terminated or truncated)
        return obs_preprocess(obs), float(rew), done, info
