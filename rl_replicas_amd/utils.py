"""Numeric utilities.

API parity with the reference's `rl_replicas.utils` (reference:
src/rl_replicas/utils.py:14-136) with MI355X-first internals: the
discounted scans and normalization dispatch to the HIP segmented-scan /
reduction kernels when tensors live on a GPU (rl_replicas_amd.ops);
numpy entry points keep the reference's exact semantics for drop-in
compatibility on CPU.
"""
from __future__ import annotations

import logging
import os
import random
from typing import Iterable, List, Optional

import numpy as np
import torch
from torch import Tensor, nn

from rl_replicas_amd.policies.policy import Policy
from rl_replicas_amd.value_function import ValueFunction

logger = logging.getLogger(__name__)


def discounted_cumulative_sums(vector: np.ndarray, discount: float) -> np.ndarray:
    """Reverse discounted cumulative sum: out[t] = sum_k discount^k * x[t+k].

    Reference semantics: src/rl_replicas/utils.py:14-28 (scipy IIR
    filter).  Implemented here as an explicit backward recurrence —
    bitwise equivalent ordering (out[t] = x[t] + discount*out[t+1]).
    The GPU hot path does not call this: whole-rollout segmented scans
    run in one HIP kernel (ops.segmented_discounted_cumsum).
    """
    x = np.asarray(vector, dtype=np.float64)
    out = np.empty_like(x)
    acc = np.zeros_like(x[0]) if x.ndim > 1 else 0.0
    for t in range(len(x) - 1, -1, -1):
        acc = x[t] + discount * acc
        out[t] = acc
    return out


def gae(rewards: np.ndarray, gamma: float, values: np.ndarray, gae_lambda: float) -> np.ndarray:
    """Generalized Advantage Estimation over one episode.

    rewards has one bootstrap element appended (length L+1), values has
    length L+1; returns L advantages.  Reference: utils.py:31-44.
    """
    rewards = np.asarray(rewards, dtype=np.float64)
    values = np.asarray(values, dtype=np.float64)
    td_residuals = rewards[:-1] + gamma * values[1:] - values[:-1]
    return discounted_cumulative_sums(td_residuals, gamma * gae_lambda)


def polyak_average(params: Iterable[nn.Parameter], target_params: Iterable[nn.Parameter], rho: float) -> None:
    """target <- rho*target + (1-rho)*param, in place, no grad.

    Reference: utils.py:47-57.  On GPU with the HIP extension loaded
    this runs as ONE fused multi-tensor lerp kernel instead of a
    per-parameter Python loop.
    """
    params = list(params)
    target_params = list(target_params)
    from rl_replicas_amd import ops

    with torch.no_grad():
        if ops.wants_hip(target_params[0]):
            ops.fused_polyak_(
                [p.data for p in params], [t.data for t in target_params], rho
            )
        else:
            for param, target_param in zip(params, target_params):
                target_param.data.mul_(rho).add_(param.data, alpha=1.0 - rho)


def compute_values(
    observations_with_last_observation: List[List[np.ndarray]],
    value_function: ValueFunction,
) -> List[np.ndarray]:
    """No-grad V(s) for each episode including the bootstrap obs.

    Reference: utils.py:60-71 runs one forward per episode; here all
    episodes are concatenated into a single batched forward (one kernel
    launch / one H2D instead of N) and split back per episode.
    """
    device = next(value_function.parameters()).device
    lengths = [len(ep) for ep in observations_with_last_observation]
    if not lengths:
        return []
    flat = np.concatenate([np.stack(ep) for ep in observations_with_last_observation], axis=0)
    with torch.no_grad():
        t = torch.as_tensor(flat, dtype=torch.float32, device=device)
        v = value_function(t).flatten().cpu().numpy()
    out: List[np.ndarray] = []
    start = 0
    for n in lengths:
        out.append(v[start : start + n])
        start += n
    return out


def bootstrap_rewards_with_last_values(
    rewards: List[List[float]], episode_dones: List[bool], last_values: List[float]
) -> List[np.ndarray]:
    """Append 0 (terminal) or V(s_last) (truncated) per episode.

    Reference: utils.py:74-87.
    """
    out: List[np.ndarray] = []
    for episode_rewards, done, last_value in zip(rewards, episode_dones, last_values):
        tail = 0.0 if done else float(last_value)
        out.append(np.asarray(list(episode_rewards) + [tail]))
    return out


def normalize_tensor(vector: Tensor) -> Tensor:
    """(x - mean) / std with Bessel-corrected std (torch default).

    Reference: utils.py:90-92.  GPU path: single-pass fused
    mean/std/apply HIP kernel (ops.normalize).
    """
    from rl_replicas_amd import ops

    return ops.normalize(vector)


def add_noise_to_get_action(policy: Policy, action_space, action_noise_scale: float) -> Policy:
    """Wrap a policy with additive clipped Gaussian exploration noise.

    Reference: utils.py:95-124 (used by DDPG/TD3 after warm-up).
    """
    return _NoisedPolicy(policy, action_space, action_noise_scale)


class _NoisedPolicy(Policy):
    def __init__(self, base_policy: Policy, action_space, action_noise_scale: float):
        super().__init__()
        self.base_policy = base_policy
        self.action_space = action_space
        self.action_noise_scale = action_noise_scale
        self.action_limit = float(np.asarray(action_space.high).reshape(-1)[0])
        self.action_size = int(action_space.shape[0])
        self._sample_offset = 0
        self._dummy_log_std: Optional[Tensor] = None

    def get_action_tensor(self, observation: Tensor) -> Tensor:
        from rl_replicas_amd import ops

        action = self.base_policy.get_action_tensor(observation)
        if (
            action.dim() == 2
            and action.dtype == torch.float32
            and ops.wants_hip(action)
        ):
            # ONE Philox kernel (noise_scale overrides sigma, limit clips)
            # instead of the randn+mul+add+clip torch chain — the DDPG/TD3
            # rollout hot path (gaussian_sample_kernel, sample_kernels.hip)
            ext = ops._load_extension()
            if self._dummy_log_std is None or self._dummy_log_std.device != action.device:
                self._dummy_log_std = torch.zeros(action.shape[1], device=action.device)
            self._sample_offset += 1
            # salt decorrelates this stream from the base policies' streams
            seed = (torch.initial_seed() ^ 0x517CC1B727220A95) & 0x7FFFFFFFFFFFFFFF
            return ext.gaussian_sample(
                action.contiguous(), self._dummy_log_std, seed,
                self._sample_offset, float(self.action_noise_scale),
                self.action_limit,
            )
        noise_shape = action.shape if action.dim() > 1 else (self.action_size,)
        action = action + self.action_noise_scale * torch.randn(
            noise_shape, device=action.device, dtype=action.dtype
        )
        return torch.clip(action, -self.action_limit, self.action_limit)

    def get_action_numpy(self, observation: np.ndarray) -> np.ndarray:
        action = self.base_policy.get_action_numpy(observation)
        action = action + self.action_noise_scale * np.random.randn(*action.shape)
        return np.clip(action, -self.action_limit, self.action_limit)


def set_seed_for_libraries(seed: int) -> None:
    """Seed random/numpy/torch and force deterministic algorithms.

    Reference: utils.py:127-136.  Determinism is the foundation of the
    test strategy.
    """
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    torch.backends.cudnn.deterministic = True
    torch.backends.cudnn.benchmark = False
    # strict on CPU (reference semantics); warn-only on ROCm where a few
    # torch eager kernels lack a deterministic variant — our own HIP
    # kernels are deterministic by construction (split-K workspace
    # reductions, no atomics) and the GPU tests assert bitwise equality
    torch.use_deterministic_algorithms(True, warn_only=torch.cuda.is_available())


def set_seed_for_rank(seed: int, rank: Optional[int] = None) -> None:
    """Distributed extension: deterministic per-rank seed offsets.

    Each data-parallel rank gets an independent sampling stream while
    the whole job stays reproducible.
    """
    if rank is None:
        rank = int(os.environ.get("RANK", "0"))
    set_seed_for_libraries(seed + 1_000_003 * rank)
