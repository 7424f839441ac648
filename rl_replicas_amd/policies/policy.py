"""Policy base class (reference: src/rl_replicas/policies/policy.py:7-30)."""
from abc import ABC, abstractmethod

import numpy as np
import torch.nn as nn
from torch import Tensor


class Policy(nn.Module, ABC):
    """Abstract policy: an `nn.Module` that maps observations to actions.

    Both entry points accept a single observation or a batch (leading
    batch dim) — the batched form is what the vectorized sampler uses.
    """

    @abstractmethod
    def get_action_tensor(self, observation: Tensor) -> Tensor:
        """Action for `observation` as a torch Tensor (no grad)."""
        raise NotImplementedError

    @abstractmethod
    def get_action_numpy(self, observation: np.ndarray) -> np.ndarray:
        """Action for `observation` as a numpy array (no grad)."""
        raise NotImplementedError
