"""Deep Deterministic Policy Gradient.

API parity: reference src/rl_replicas/algorithms/ddpg.py:25-314 —
deterministic actor + single critic, frozen deepcopied targets,
Q-target r + gamma*(1-d)*Q'(s', mu'(s')), actor loss -E[Q(s, mu(s))]
with the critic frozen during the actor step, Polyak rho=0.995 on both
targets every train iteration.
"""
from __future__ import annotations

from typing import Dict, List

import copy
import numpy as np
import torch
from torch import Tensor

from rl_replicas_amd import ops
from rl_replicas_amd.algorithms.off_policy import OffPolicyAlgorithm
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.q_function import QFunction
from rl_replicas_amd.replay_buffer import ReplayBuffer
from rl_replicas_amd.samplers import Sampler
from rl_replicas_amd.utils import polyak_average


class DDPG(OffPolicyAlgorithm):
    def __init__(
        self,
        policy: Policy,
        exploration_policy: Policy,
        q_function: QFunction,
        env,
        sampler: Sampler,
        replay_buffer: ReplayBuffer,
        evaluator: Evaluator,
        gamma: float = 0.99,
        polyak_rho: float = 0.995,
        action_noise_scale: float = 0.1,
    ) -> None:
        self.q_function = q_function
        super().__init__(
            policy,
            exploration_policy,
            env,
            sampler,
            replay_buffer,
            evaluator,
            gamma,
            polyak_rho,
            action_noise_scale,
        )
        self.target_q_function = copy.deepcopy(self.q_function)
        for param in self.target_q_function.network.parameters():
            param.requires_grad = False

    # ------------------------------------------------------------------
    def train(self, replay_buffer: ReplayBuffer, num_train_steps: int, minibatch_size: int) -> None:
        """50-iteration minibatch loop (reference ddpg.py:195-253).

        Losses/Q-values stay device-resident across the loop (one
        readback at the end); on GPU the Q and actor steps run through
        the fused kernel path (ops.fused_offpolicy)."""
        from rl_replicas_amd.ops import fused_offpolicy as fop

        if fop.graph_supported(self, minibatch_size):
            # single-process GPU: whole loop = one captured hipGraph
            self._record_offpolicy_metrics(
                fop.graphed_epoch(self, num_train_steps, minibatch_size)
            )
            return

        policy_losses: List[Tensor] = []
        q_losses: List[Tensor] = []
        all_q_values: List[Tensor] = []

        for _ in range(num_train_steps):
            mb = self._sample_minibatch_device(minibatch_size)
            observations = mb["observations"]
            actions = mb["actions"]
            fused = fop.supported(self.q_function, observations) and fop.supported(
                self.policy, observations
            )

            with torch.no_grad():
                all_q_values.append(self.q_function(observations, actions))

            targets = self.compute_targets(mb["next_observations"], mb["rewards"], mb["dones"])
            if fused:
                q_losses.append(
                    fop.q_step(self.q_function, observations, actions, targets,
                               self._all_reduce_gradients)
                )
                policy_losses.append(
                    fop.policy_step(self.policy, self.q_function, observations,
                                    self._all_reduce_gradients)
                )
            else:
                q_losses.append(
                    self._train_q_single(self.q_function, observations, actions, targets)
                )
                policy_losses.append(self.train_policy(observations))

            polyak_average(
                self.policy.network.parameters(),
                self.target_policy.network.parameters(),
                self.polyak_rho,
            )
            polyak_average(
                self.q_function.network.parameters(),
                self.target_q_function.network.parameters(),
                self.polyak_rho,
            )

        q_values = torch.cat(all_q_values)
        m = self.metrics_manager
        m.record_scalar(
            "policy/average_loss",
            float(torch.stack(policy_losses).mean()),
            self.current_total_steps,
            tensorboard=True,
        )
        m.record_scalar(
            "q-function/average_loss",
            float(torch.stack(q_losses).mean()),
            self.current_total_steps,
            tensorboard=True,
        )
        m.record_scalar(
            "q-function/avarage_q-value", float(q_values.mean()), self.current_total_steps, tensorboard=True
        )
        m.record_scalar("q-function/max_q-value", float(q_values.max()))
        m.record_scalar("q-function/min_q-value", float(q_values.min()))

    # ------------------------------------------------------------------
    def train_policy(self, observations: Tensor) -> Tensor:
        """Actor step with critic frozen (reference ddpg.py:255-273)."""
        for param in self.q_function.network.parameters():
            param.requires_grad = False

        policy_actions = self.policy(observations)
        loss = -torch.mean(self.q_function(observations, policy_actions))
        self.policy.optimizer.zero_grad()
        loss.backward()
        self._all_reduce_gradients(self.policy)
        self.policy.optimizer.step()

        for param in self.q_function.network.parameters():
            param.requires_grad = True
        return loss.detach()

    def compute_targets(self, next_observations: Tensor, rewards: Tensor, dones: Tensor) -> Tensor:
        """r + gamma*(1-d)*Q'(s', mu'(s'))  (reference ddpg.py:275-282)."""
        with torch.no_grad():
            next_actions = self.target_policy(next_observations)
            target_q = self.target_q_function(next_observations, next_actions)
        return ops.q_target(rewards, dones.float(), target_q, self.gamma)

    # ------------------------------------------------------------------
    def _checkpoint_dict(self, epoch: int) -> Dict:
        return {
            "epoch": epoch,
            "total_steps": self.current_total_steps,
            "policy_state_dict": self.policy.network.state_dict(),
            "policy_optimizer_state_dict": self.policy.optimizer.state_dict(),
            "target_policy_state_dict": self.target_policy.network.state_dict(),
            "q_function_state_dict": self.q_function.network.state_dict(),
            "q_function_optimizer_state_dict": self.q_function.optimizer.state_dict(),
            "target_q_function_state_dict": self.target_q_function.network.state_dict(),
        }

    def _restore_from_checkpoint(self, ckpt: Dict) -> None:
        self.policy.network.load_state_dict(ckpt["policy_state_dict"])
        self.policy.optimizer.load_state_dict(ckpt["policy_optimizer_state_dict"])
        self.target_policy.network.load_state_dict(ckpt["target_policy_state_dict"])
        self.q_function.network.load_state_dict(ckpt["q_function_state_dict"])
        self.q_function.optimizer.load_state_dict(ckpt["q_function_optimizer_state_dict"])
        self.target_q_function.network.load_state_dict(ckpt["target_q_function_state_dict"])
