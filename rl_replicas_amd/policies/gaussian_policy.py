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
        self._sample_offset = 0

    def forward(self, observation: Tensor) -> Independent:
        mean: Tensor = self.network(observation)
        std = torch.exp(self.log_std)
        return Independent(Normal(mean, std), 1)

    def get_action_tensor(self, observation: Tensor) -> Tensor:
        # GPU fast path: fused MLP forward + one Philox sample kernel
        # instead of the exp/randn/mul/add torch chain (rollout hot loop)
        from rl_replicas_amd import ops

        if observation.dim() == 2 and ops.wants_hip(observation):
            from rl_replicas_amd.ops.fused_mlp import _extract_layers

            if _extract_layers(self.network) is not None:
                with torch.no_grad():
                    mean = self.network(observation)
                    ext = ops._load_extension()
                    self._sample_offset += 1
                    return ext.gaussian_sample(
                        mean,
                        self.log_std.data,
                        torch.initial_seed() & 0x7FFFFFFFFFFFFFFF,
                        self._sample_offset,
                        -1.0,
                        -1.0,
                    )
        return super().get_action_tensor(observation)
