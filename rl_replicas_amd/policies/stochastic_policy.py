"""Stochastic policy ABC (reference: src/rl_replicas/policies/stochastic_policy.py:11-41)."""
from abc import abstractmethod

import numpy as np
import torch
from torch import Tensor
from torch.distributions import Distribution

from .policy import Policy


class StochasticPolicy(Policy):
    """Policy whose forward returns a torch Distribution; actions are samples."""

    @abstractmethod
    def forward(self, observation: Tensor) -> Distribution:
        raise NotImplementedError

    def get_action_tensor(self, observation: Tensor) -> Tensor:
        with torch.no_grad():
            return self(observation).sample()

    def get_action_numpy(self, observation: np.ndarray) -> np.ndarray:
        device = next(self.parameters()).device
        obs = torch.as_tensor(np.asarray(observation), dtype=torch.float32, device=device)
        return self.get_action_tensor(obs).cpu().numpy()
