"""Twin Delayed DDPG (TD3).

API parity: reference src/rl_replicas/algorithms/td3.py:25-382 — twin
critics, target-policy smoothing (clipped Gaussian noise on the target
action, clamped to the action limit), min(Q1', Q2') targets, delayed
actor/target updates every `policy_delay` steps.
"""
from __future__ import annotations

import copy
from typing import Dict, List

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


class TD3(OffPolicyAlgorithm):
    def __init__(
        self,
        policy: Policy,
        exploration_policy: Policy,
        q_function_1: QFunction,
        q_function_2: QFunction,
        env,
        sampler: Sampler,
        replay_buffer: ReplayBuffer,
        evaluator: Evaluator,
        gamma: float = 0.99,
        polyak_rho: float = 0.995,
        action_noise_scale: float = 0.1,
        target_noise_scale: float = 0.2,
        target_noise_clip: float = 0.5,
        policy_delay: int = 2,
    ) -> None:
        self.q_function_1 = q_function_1
        self.q_function_2 = q_function_2
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
        self.target_noise_scale = target_noise_scale
        self.target_noise_clip = target_noise_clip
        self.policy_delay = policy_delay
        self.target_q_function_1 = copy.deepcopy(self.q_function_1)
        self.target_q_function_2 = copy.deepcopy(self.q_function_2)
        for q in (self.target_q_function_1, self.target_q_function_2):
            for param in q.network.parameters():
                param.requires_grad = False
        self._smooth_offset = 0  # Philox offset for the GPU smoothing kernel

    # ------------------------------------------------------------------
    def train(self, replay_buffer: ReplayBuffer, num_train_steps: int, minibatch_size: int) -> None:
        """Twin-critic minibatch loop (reference td3.py:214-263); losses
        stay device-resident, fused kernel steps on GPU."""
        from rl_replicas_amd.ops import fused_offpolicy as fop

        if fop.graph_supported(self, minibatch_size):
            # single-process GPU: the whole 50-iteration loop is ONE
            # captured hipGraph replay (fused_offpolicy._GraphedOffPolicy)
            self._record_offpolicy_metrics(
                fop.graphed_epoch(self, num_train_steps, minibatch_size)
            )
            return

        policy_losses: List[Tensor] = []
        q1_losses: List[Tensor] = []
        q2_losses: List[Tensor] = []
        all_q1: List[Tensor] = []
        all_q2: List[Tensor] = []

        for train_step in range(num_train_steps):
            mb = self._sample_minibatch_device(minibatch_size)
            observations = mb["observations"]
            actions = mb["actions"]
            fused = fop.supported(self.q_function_1, observations) and fop.supported(
                self.policy, observations
            )

            with torch.no_grad():
                all_q1.append(self.q_function_1(observations, actions))
                all_q2.append(self.q_function_2(observations, actions))

            targets = self.compute_targets(mb["next_observations"], mb["rewards"], mb["dones"])
            if fused:
                q1_losses.append(
                    fop.q_step(self.q_function_1, observations, actions, targets,
                               self._all_reduce_gradients)
                )
                q2_losses.append(
                    fop.q_step(self.q_function_2, observations, actions, targets,
                               self._all_reduce_gradients)
                )
            else:
                q1_losses.append(
                    self._train_q_single(self.q_function_1, observations, actions, targets)
                )
                q2_losses.append(
                    self._train_q_single(self.q_function_2, observations, actions, targets)
                )

            if train_step % self.policy_delay == 0:
                if fused:
                    policy_losses.append(
                        fop.policy_step(self.policy, self.q_function_1, observations,
                                        self._all_reduce_gradients)
                    )
                else:
                    policy_losses.append(self.train_policy(observations))
                polyak_average(
                    self.policy.network.parameters(),
                    self.target_policy.network.parameters(),
                    self.polyak_rho,
                )
                polyak_average(
                    self.q_function_1.network.parameters(),
                    self.target_q_function_1.network.parameters(),
                    self.polyak_rho,
                )
                polyak_average(
                    self.q_function_2.network.parameters(),
                    self.target_q_function_2.network.parameters(),
                    self.polyak_rho,
                )

        q1 = torch.cat(all_q1)
        q2 = torch.cat(all_q2)
        m = self.metrics_manager
        m.record_scalar(
            "policy/average_loss",
            float(torch.stack(policy_losses).mean()),
            self.current_total_steps,
            tensorboard=True,
        )
        m.record_scalar(
            "q-function_1/average_loss",
            float(torch.stack(q1_losses).mean()),
            self.current_total_steps,
            tensorboard=True,
        )
        m.record_scalar(
            "q-function_2/average_loss",
            float(torch.stack(q2_losses).mean()),
            self.current_total_steps,
            tensorboard=True,
        )
        m.record_scalar(
            "q-function_1/avarage_q-value", float(q1.mean()), self.current_total_steps, tensorboard=True
        )
        m.record_scalar("q-function_1/max_q-value", float(q1.max()))
        m.record_scalar("q-function_1/min_q-value", float(q1.min()))
        m.record_scalar(
            "q-function_2/avarage_q-value", float(q2.mean()), self.current_total_steps, tensorboard=True
        )
        m.record_scalar("q-function_2/max_q-value", float(q2.max()))
        m.record_scalar("q-function_2/min_q-value", float(q2.min()))

    # ------------------------------------------------------------------
    def train_policy(self, observations: Tensor) -> Tensor:
        """Actor step through Q1 with both critics frozen
        (reference td3.py:300-323)."""
        for q in (self.q_function_1, self.q_function_2):
            for param in q.network.parameters():
                param.requires_grad = False

        policy_actions = self.policy(observations)
        loss = -torch.mean(self.q_function_1(observations, policy_actions))
        self.policy.optimizer.zero_grad()
        loss.backward()
        self._all_reduce_gradients(self.policy)
        self.policy.optimizer.step()

        for q in (self.q_function_1, self.q_function_2):
            for param in q.network.parameters():
                param.requires_grad = True
        return loss.detach()

    def compute_targets(self, next_observations: Tensor, rewards: Tensor, dones: Tensor) -> Tensor:
        """Smoothed min-twin target (reference td3.py:325-341).

        GPU: smoothing noise+clip+limit is ONE Philox kernel
        (ops.td3_smooth) and the min-twin bootstrap is one fused kernel
        (ops.q_target_min2); CPU keeps the reference's torch.randn RNG
        stream (tests/test_reference_equivalence.py)."""
        from rl_replicas_amd.ops import fused_offpolicy as fop

        with torch.no_grad():
            next_actions = self.target_policy(next_observations)
            action_limit = float(np.asarray(self.env.action_space.high).reshape(-1)[0])
            if ops.wants_hip(next_actions):
                self._smooth_offset += 1
                next_actions = ops.td3_smooth(
                    next_actions, self.target_noise_scale,
                    self.target_noise_clip, action_limit,
                    fop._stream_seed(), self._smooth_offset,
                )
            else:
                next_actions = ops.td3_smooth(
                    next_actions, self.target_noise_scale,
                    self.target_noise_clip, action_limit, 0, 0,
                )
            q1 = self.target_q_function_1(next_observations, next_actions)
            q2 = self.target_q_function_2(next_observations, next_actions)
        return ops.q_target_min2(rewards, dones.float(), q1, q2, self.gamma)

    # ------------------------------------------------------------------
    def _checkpoint_dict(self, epoch: int) -> Dict:
        return {
            "epoch": epoch,
            "total_steps": self.current_total_steps,
            "policy_state_dict": self.policy.network.state_dict(),
            "policy_optimizer_state_dict": self.policy.optimizer.state_dict(),
            "target_policy_state_dict": self.target_policy.network.state_dict(),
            "q_function_1_state_dict": self.q_function_1.network.state_dict(),
            "q_function_1_optimizer_state_dict": self.q_function_1.optimizer.state_dict(),
            "target_q_function_1_state_dict": self.target_q_function_1.network.state_dict(),
            "q_function_2_state_dict": self.q_function_2.network.state_dict(),
            "q_function_2_optimizer_state_dict": self.q_function_2.optimizer.state_dict(),
            "target_q_function_2_state_dict": self.target_q_function_2.network.state_dict(),
        }

    def _restore_from_checkpoint(self, ckpt: Dict) -> None:
        self.policy.network.load_state_dict(ckpt["policy_state_dict"])
        self.policy.optimizer.load_state_dict(ckpt["policy_optimizer_state_dict"])
        self.target_policy.network.load_state_dict(ckpt["target_policy_state_dict"])
        for i, (q, tq) in enumerate(
            [
                (self.q_function_1, self.target_q_function_1),
                (self.q_function_2, self.target_q_function_2),
            ],
            start=1,
        ):
            q.network.load_state_dict(ckpt[f"q_function_{i}_state_dict"])
            q.optimizer.load_state_dict(ckpt[f"q_function_{i}_optimizer_state_dict"])
            tq.network.load_state_dict(ckpt[f"target_q_function_{i}_state_dict"])
