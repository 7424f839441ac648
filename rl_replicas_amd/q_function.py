"""Action-value function Q(s, a) (reference: src/rl_replicas/q_function.py:6-32)."""
import torch
import torch.nn as nn
from torch import Tensor
from torch.optim import Optimizer


class QFunction(nn.Module):
    """Q(s, a) approximator over concatenated [obs, action]; owns its optimizer."""

    def __init__(self, network: nn.Module, optimizer: Optimizer) -> None:
        super().__init__()
        self.network = network
        self.optimizer = optimizer

    def forward(self, observation: Tensor, action: Tensor) -> Tensor:
        q: Tensor = self.network(torch.cat([observation, action], dim=-1))
        return torch.squeeze(q, dim=-1)
