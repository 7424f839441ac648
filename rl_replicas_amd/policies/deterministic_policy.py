"""Deterministic policy (DDPG/TD3 actor).

Reference: src/rl_replicas/policies/deterministic_policy.py:9-45.
"""
import numpy as np
import torch
import torch.nn as nn
from torch import Tensor
from torch.optim import Optimizer

from .policy import Policy


class DeterministicPolicy(Policy):
    def __init__(self, network: nn.Module, optimizer: Optimizer):
        super().__init__()
        self.network = network
        self.optimizer = optimizer

    def forward(self, observation: Tensor) -> Tensor:
        return self.network(observation)

    def get_action_tensor(self, observation: Tensor) -> Tensor:
        with torch.no_grad():
            return self(observation)

    def get_action_numpy(self, observation: np.ndarray) -> np.ndarray:
        device = next(self.parameters()).device
        obs = torch.as_tensor(np.asarray(observation), dtype=torch.float32, device=device)
        return self.get_action_tensor(obs).cpu().numpy()
