"""Proximal Policy Optimization (clipped surrogate).

API parity: reference src/rl_replicas/algorithms/ppo.py:30-306 —
clipped-ratio loss with epsilon=`clip_range`, up to
`num_policy_gradients` Adam steps per epoch with early stop when the
approximate KL (mean(old_logp - logp)) exceeds 1.5*`max_kl_divergence`,
old policy snapshot synced after the epoch.

This is the framework's flagship/benchmark algorithm
(BASELINE.json config #2/#5: PPO HalfCheetah-v4 on 1-8 MI355X).
"""
from __future__ import annotations

import copy
import logging
from typing import Dict

import torch
from torch import Tensor

from rl_replicas_amd.algorithms.on_policy import OnPolicyAlgorithm
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.samplers import Sampler
from rl_replicas_amd.value_function import ValueFunction

logger = logging.getLogger(__name__)


class PPO(OnPolicyAlgorithm):
    def __init__(
        self,
        policy: Policy,
        value_function: ValueFunction,
        env,
        sampler: Sampler,
        gamma: float = 0.99,
        gae_lambda: float = 0.97,
        clip_range: float = 0.2,
        max_kl_divergence: float = 0.01,
        num_policy_gradients: int = 80,
        num_value_gradients: int = 80,
    ) -> None:
        super().__init__(policy, value_function, env, sampler, gamma, gae_lambda, num_value_gradients)
        self.clip_range = clip_range
        self.max_kl_divergence = max_kl_divergence
        self.num_policy_gradients = num_policy_gradients
        self.old_policy: Policy = copy.deepcopy(self.policy)

    # ------------------------------------------------------------------
    def _update_policy(self, obs: Tensor, actions: Tensor, advantages: Tensor) -> Dict[str, float]:
        from rl_replicas_amd.ops import fused_onpolicy

        if fused_onpolicy.supported(self.policy, obs):
            return fused_onpolicy.ppo_update(self, obs, actions, advantages)
        # pre-update diagnostics (reference ppo.py:163-170)
        diagnostics = self._policy_diagnostics(obs, actions)
        with torch.no_grad():
            loss_before = self.compute_policy_loss(obs, actions, advantages)

        # old_logp is constant across the whole update loop: compute once
        with torch.no_grad():
            old_log_probs = self.old_policy(obs).log_prob(actions)

        approximate_kl = torch.zeros((), device=obs.device)
        for i in range(self.num_policy_gradients):
            self.train_policy(obs, actions, advantages, old_log_probs)
            with torch.no_grad():
                log_probs = self.policy(obs).log_prob(actions)
                approximate_kl = torch.mean(old_log_probs - log_probs)
            # DP: all ranks must agree on the early stop -> reduce the KL
            approximate_kl = self._reduce_scalar_mean(approximate_kl)
            if float(approximate_kl) > 1.5 * self.max_kl_divergence:
                logger.info(
                    "Early stopping at update %d due to reaching max KL divergence.", i
                )
                break

        self.old_policy.load_state_dict(self.policy.state_dict())

        return {
            "policy/loss": float(loss_before),
            **diagnostics,
            "policy/kl_divergence": float(approximate_kl),
        }

    def train_policy(self, obs: Tensor, actions: Tensor, advantages: Tensor, old_log_probs: Tensor) -> None:
        loss = self._clipped_loss(obs, actions, advantages, old_log_probs)
        self.policy.optimizer.zero_grad()
        loss.backward()
        self._all_reduce_gradients(self.policy)
        self.policy.optimizer.step()

    def _clipped_loss(self, obs: Tensor, actions: Tensor, advantages: Tensor, old_log_probs: Tensor) -> Tensor:
        """-E[min(r*A, clip(r, 1-eps, 1+eps)*A)]  (reference ppo.py:237-257)."""
        log_probs = self.policy(obs).log_prob(actions)
        ratio = torch.exp(log_probs - old_log_probs)
        clipped_ratio = torch.clamp(ratio, 1.0 - self.clip_range, 1.0 + self.clip_range)
        return -torch.mean(torch.min(ratio * advantages, clipped_ratio * advantages))

    # reference-compatible helpers -------------------------------------
    def compute_policy_loss(self, observations: Tensor, actions: Tensor, advantages: Tensor) -> Tensor:
        with torch.no_grad():
            old_log_probs = self.old_policy(observations).log_prob(actions)
        return self._clipped_loss(observations, actions, advantages, old_log_probs)

    def compute_approximate_kl_divergence(self, observations: Tensor, actions: Tensor) -> Tensor:
        """mean(old_logp - logp)  (reference ppo.py:259-269)."""
        with torch.no_grad():
            log_probs = self.policy(observations).log_prob(actions)
            old_log_probs = self.old_policy(observations).log_prob(actions)
        return torch.mean(old_log_probs - log_probs)
