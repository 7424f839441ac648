"""Diagonal Gaussian policy with state-independent learnable log-std.

Reference: src/rl_replicas/policies/gaussian_policy.py:9-37 — the
`log_std` nn.Parameter is passed in by the user and the distribution is
`Independent(Normal(mean, exp(log_std)), 1)`.
"""
import torch
import torch.nn as nn
from torch import Tensor
from torch.distributions import Independent, Normal
from torch.optim import Optimizer

from .stochastic_policy import StochasticPolicy


class GaussianPolicy(StochasticPolicy):
    def __init__(self, network: nn.Module, optimizer: Optimizer, log_std: nn.Parameter):
        super().__init__()
        self.network = network
        self.optimizer = optimizer
        self.log_std = log_std

    def forward(self, observation: Tensor) -> Independent:
        mean: Tensor = self.network(observation)
        std = torch.exp(self.log_std)
        return Independent(Normal(mean, std), 1)
