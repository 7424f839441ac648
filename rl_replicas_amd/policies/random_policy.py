"""Uniform-random policy over an action space (exploration warm-up).

Reference: src/rl_replicas/policies/random_policy.py:9-27 (used by
DDPG/TD3 for the first `num_start_steps`).  Supports batched
observations: returns one independent sample per row.
"""
import numpy as np
import torch
from torch import Tensor

from .policy import Policy


class RandomPolicy(Policy):
    def __init__(self, action_space):
        super().__init__()
        self.action_space = action_space

    def _sample(self, observation_shape) -> np.ndarray:
        # batched obs [B, D] -> B independent samples (one vectorized draw
        # when the space supports it)
        if len(observation_shape) > 1:
            batch = observation_shape[0]
            try:
                return np.asarray(self.action_space.sample(batch))
            except TypeError:  # third-party space without batch support
                return np.stack([self.action_space.sample() for _ in range(batch)])
        return np.asarray(self.action_space.sample())

    def get_action_tensor(self, observation: Tensor) -> Tensor:
        return torch.as_tensor(self._sample(tuple(observation.shape)))

    def get_action_numpy(self, observation: np.ndarray) -> np.ndarray:
        return self._sample(np.asarray(observation).shape)
