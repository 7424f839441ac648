"""Trust Region Policy Optimization.

API parity: reference src/rl_replicas/algorithms/trpo.py:30-277 —
surrogate loss -E[exp(logp - old_logp)*A], KL constraint
mean(KL(old || new)), one trust-region step per epoch via
ConjugateGradientOptimizer (CG on the Fisher, backtracking line
search), old policy deepcopy synced after each update.

MI355X note: the FVP needs double backward through the policy forward,
which the fused-MLP training kernels do not support — TRPO pins the
policy's MLPs to the eager (rocBLAS) path for grad-enabled forwards
(`fused_training = False`); no_grad forwards (sampling/eval) still use
the fused kernel.
"""
from __future__ import annotations

from typing import Callable, Dict

import torch
from torch import Tensor
from torch.distributions import kl_divergence

from rl_replicas_amd.algorithms.on_policy import OnPolicyAlgorithm
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.samplers import Sampler
from rl_replicas_amd.value_function import ValueFunction

import copy


class TRPO(OnPolicyAlgorithm):
    def __init__(
        self,
        policy: Policy,
        value_function: ValueFunction,
        env,
        sampler: Sampler,
        gamma: float = 0.99,
        gae_lambda: float = 0.97,
        num_value_gradients: int = 80,
    ) -> None:
        super().__init__(policy, value_function, env, sampler, gamma, gae_lambda, num_value_gradients)
        self.old_policy: Policy = copy.deepcopy(self.policy)
        # FVP requires double backward -> force eager path when grads are on
        from rl_replicas_amd.networks import MLP

        for module in self.policy.modules():
            if isinstance(module, MLP):
                module.fused_training = False

    # ------------------------------------------------------------------
    def _update_policy(self, obs: Tensor, actions: Tensor, advantages: Tensor) -> Dict[str, float]:
        def compute_surrogate_loss() -> Tensor:
            log_probs = self.policy(obs).log_prob(actions)
            with torch.no_grad():
                old_log_probs = self.old_policy(obs).log_prob(actions)
            ratio = torch.exp(log_probs - old_log_probs)
            return -torch.mean(ratio * advantages)

        def compute_kl_constraint() -> Tensor:
            dist = self.policy(obs)
            with torch.no_grad():
                old_dist = self.old_policy(obs)
            return torch.mean(kl_divergence(old_dist, dist))

        diagnostics = self._policy_diagnostics(obs, actions)
        loss_before = compute_surrogate_loss()

        self.train_policy(compute_surrogate_loss, compute_kl_constraint, obs)

        self.old_policy.load_state_dict(self.policy.state_dict())

        return {"policy/loss": float(loss_before.detach()), **diagnostics}

    def train_policy(
        self,
        compute_surrogate_loss: Callable,
        compute_kl_constraint: Callable,
        observations: Tensor = None,
    ) -> None:
        """Populate loss grads, then run the CG trust-region step
        (reference trpo.py:228-240).

        On GPU the CG solve uses the analytic Fisher-vector product
        (ops/fused_trpo.py) — exact here because the KL is evaluated at
        policy == old_policy — instead of double backward."""
        loss = compute_surrogate_loss()
        self.policy.optimizer.zero_grad()
        loss.backward()
        self._all_reduce_gradients(self.policy)

        fvp = None
        if observations is not None and observations.is_cuda:
            from rl_replicas_amd.ops import fused_trpo

            fvp = fused_trpo.make_fvp(
                self.policy, observations,
                self.policy.optimizer.hvp_damping_coefficient,
            )
        # under all-reduce DP every CG iteration and line-search
        # evaluation must see the GLOBAL batch (mean over ranks) or
        # replicas diverge; under replicate DP the batch IS global
        # locally, so no hook (and the captured CG solve stays active)
        reduce_hook = (
            self._reduce_scalar_mean
            if getattr(self, "_dp_enabled", False)
            and not getattr(self, "_dp_replicate", False)
            else None
        )
        self.policy.optimizer.step(
            compute_surrogate_loss, compute_kl_constraint,
            fisher_vector_product=fvp, reduce_hook=reduce_hook,
        )
