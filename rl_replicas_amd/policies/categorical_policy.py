"""Categorical policy for discrete action spaces.

Reference: src/rl_replicas/policies/categorical_policy.py:8-32.  Like
every function approximator in this library, the policy owns its
optimizer (constructor-injection pattern).
"""
import torch.nn as nn
from torch import Tensor
from torch.distributions import Categorical
from torch.optim import Optimizer

from .stochastic_policy import StochasticPolicy


class CategoricalPolicy(StochasticPolicy):
    def __init__(self, network: nn.Module, optimizer: Optimizer):
        super().__init__()
        self.network = network
        self.optimizer = optimizer

    def forward(self, observation: Tensor) -> Categorical:
        logits: Tensor = self.network(observation)
        return Categorical(logits=logits)
