"""State-value function V(s) (reference: src/rl_replicas/value_function.py:5-28)."""
import torch.nn as nn
from torch import Tensor
from torch.optim import Optimizer


class ValueFunction(nn.Module):
    """V(s) approximator; owns its optimizer (constructor injection)."""

    def __init__(self, network: nn.Module, optimizer: Optimizer) -> None:
        super().__init__()
        self.network = network
        self.optimizer = optimizer

    def forward(self, observation: Tensor) -> Tensor:
        return self.network(observation)
