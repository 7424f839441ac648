"""MLP function approximator.

API parity: reference src/rl_replicas/networks/mlp.py:6-41 (`MLP(sizes,
activation_function, output_activation_function)` building an
`nn.Sequential` named `network`, so checkpoints keep the exact
`network.{0,2,4}.{weight,bias}` tensor naming — SURVEY.md §5.4).

MI355X fast path: when the input lives on a GPU and the HIP extension
is available, `forward` runs the whole multi-layer forward in a single
fused CDNA4 kernel (weights staged in LDS, MFMA GEMMs, activation
fused into the epilogue — rl_replicas_amd.ops.mlp) with a custom
autograd backward producing dgrad+wgrad from fused kernels.  The
`nn.Sequential` path remains the CPU / fallback implementation and the
numerics oracle for kernel tests.
"""
from __future__ import annotations

from typing import List, Type

import torch.nn as nn
from torch import Tensor


class MLP(nn.Module):
    def __init__(
        self,
        sizes: List[int],
        activation_function: Type[nn.Module] = nn.Tanh,
        output_activation_function: Type[nn.Module] = nn.Identity,
    ) -> None:
        super().__init__()
        self.sizes = list(sizes)
        self.activation_function = activation_function
        self.output_activation_function = output_activation_function

        layers: List[nn.Module] = []
        n_layers = len(sizes) - 1
        for i in range(n_layers):
            act = activation_function if i < n_layers - 1 else output_activation_function
            layers.append(nn.Linear(sizes[i], sizes[i + 1]))
            layers.append(act())
        self.network: nn.Module = nn.Sequential(*layers)

    def forward(self, input: Tensor) -> Tensor:
        from rl_replicas_amd import ops

        if ops.wants_hip(input):
            fused = ops.mlp_fused_forward(self, input)
            if fused is not NotImplemented:
                return fused
        return self.network(input)
