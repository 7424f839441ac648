"""Off-policy algorithm template (DDPG / TD3).

Reference structure (ddpg.py:75-253, td3.py:94-299): per epoch, sample
`batch_size` env steps with the exploration/noised policy through a
continuous sampler, push into the replay buffer, then (once warm) run
`num_train_steps` minibatch updates; evaluate periodically in a
separate env.

MI355X redesign: the replay buffer is an HBM-resident ring when the
networks live on GPU — minibatch gather, Q-target computation
(ops.q_target fused kernel incl. TD3 smoothing noise path), the MSE
steps and the fused Polyak multi-tensor lerp all run on device; the
only host traffic per train step is the loss scalars, which are read
back once per epoch.
"""
from __future__ import annotations

import copy
import logging
from typing import Dict

import numpy as np
import torch
from torch import Tensor
import torch.nn.functional as F

from rl_replicas_amd import envs as _envs
from rl_replicas_amd.algorithms.base import AlgorithmBase
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.replay_buffer import ReplayBuffer
from rl_replicas_amd.samplers import Sampler
from rl_replicas_amd.utils import add_noise_to_get_action

logger = logging.getLogger(__name__)


class OffPolicyAlgorithm(AlgorithmBase):
    def __init__(
        self,
        policy: Policy,
        exploration_policy: Policy,
        env,
        sampler: Sampler,
        replay_buffer: ReplayBuffer,
        evaluator: Evaluator,
        gamma: float,
        polyak_rho: float,
        action_noise_scale: float,
    ) -> None:
        self.policy = policy
        self.exploration_policy = exploration_policy
        self.env = env
        self.sampler = sampler
        self.replay_buffer = replay_buffer
        self.evaluator = evaluator
        self.gamma = gamma
        self.polyak_rho = polyak_rho
        self.action_noise_scale = action_noise_scale

        self.noised_policy = add_noise_to_get_action(
            self.policy, self.env.action_space, self.action_noise_scale
        )
        self.evaluation_env = _envs.make(env.spec.id)
        self.target_policy = copy.deepcopy(self.policy)
        for param in self.target_policy.network.parameters():
            param.requires_grad = False

    # ------------------------------------------------------------------
    def _record_offpolicy_metrics(self, metrics: Dict[str, float]) -> None:
        """Record a graphed-epoch metric dict with the reference's
        tensorboard flags (ddpg.py:234-253, td3.py:265-299)."""
        m = self.metrics_manager
        for tag, value in metrics.items():
            to_tb = tag.endswith("average_loss") or tag.endswith("avarage_q-value")
            if to_tb:
                m.record_scalar(tag, value, self.current_total_steps, tensorboard=True)
            else:
                m.record_scalar(tag, value)

    # ------------------------------------------------------------------
    def learn(
        self,
        num_epochs: int = 2000,
        batch_size: int = 50,
        minibatch_size: int = 100,
        num_start_steps: int = 10000,
        num_steps_before_update: int = 1000,
        num_train_steps: int = 50,
        num_evaluation_episodes: int = 5,
        evaluation_interval: int = 4000,
        model_saving_interval: int = 4000,
        output_dir: str = ".",
    ) -> None:
        import time as _time

        self._begin_learn(output_dir)
        for current_epoch in range(1, num_epochs + 1):
            behavior = (
                self.exploration_policy
                if self.current_total_steps < num_start_steps
                else self.noised_policy
            )
            t0 = _time.perf_counter()
            experience: Experience = self.sampler.sample(batch_size, behavior)
            self.replay_buffer.add_experience(experience)
            t1 = _time.perf_counter()

            self.current_total_steps += sum(experience.episode_lengths)
            self.current_total_episodes += sum(experience.flattened_dones)

            self._record_sampling_metrics(
                current_epoch, experience.episode_returns, experience.episode_lengths
            )

            if self.current_total_steps >= num_steps_before_update:
                self.train(self.replay_buffer, num_train_steps, minibatch_size)
            self.metrics_manager.record_phase_ms("sample", (t1 - t0) * 1000.0)
            self.metrics_manager.record_phase_ms("train", (_time.perf_counter() - t1) * 1000.0)

            if (
                num_evaluation_episodes > 0
                and self.current_total_steps % evaluation_interval == 0
            ):
                t2 = _time.perf_counter()
                self._run_evaluation(num_evaluation_episodes)
                self.metrics_manager.record_phase_ms("evaluate", (_time.perf_counter() - t2) * 1000.0)
            self.metrics_manager.dump_phases(self.current_total_steps)

            self._end_epoch(current_epoch, model_saving_interval, output_dir)
        self.metrics_manager.close()

    def _run_evaluation(self, num_evaluation_episodes: int) -> None:
        returns, lengths = self.evaluator.evaluate(
            self.policy, self.evaluation_env, num_evaluation_episodes
        )
        m = self.metrics_manager
        m.record_scalar(
            "evaluation/average_episode_return",
            float(np.mean(returns)),
            self.current_total_steps,
            tensorboard=True,
        )
        m.record_scalar("evaluation/episode_return_std", float(np.std(returns)))
        m.record_scalar("evaluation/max_episode_return", float(np.max(returns)))
        m.record_scalar("evaluation/min_episode_return", float(np.min(returns)))
        m.record_scalar(
            "evaluation/average_episode_length",
            float(np.mean(lengths)),
            self.current_total_steps,
            tensorboard=True,
        )

    # ------------------------------------------------------------------
    def _sample_minibatch_device(self, minibatch_size: int) -> Dict[str, Tensor]:
        """Minibatch on the compute device (no host round-trip on GPU)."""
        mb = self.replay_buffer.sample_minibatch_tensors(minibatch_size)
        device = self.device
        return {k: v.to(device) for k, v in mb.items()}

    def train(self, replay_buffer: ReplayBuffer, num_train_steps: int, minibatch_size: int) -> None:
        raise NotImplementedError

    def _train_q_single(self, q_function, observations: Tensor, actions: Tensor, targets: Tensor) -> Tensor:
        q_values = q_function(observations, actions)
        loss = F.mse_loss(q_values, targets)
        q_function.optimizer.zero_grad()
        loss.backward()
        self._all_reduce_gradients(q_function)
        q_function.optimizer.step()
        return loss.detach()

    def _all_reduce_gradients(self, module) -> None:
        """DP hook: no-op single-process; the parallel wrapper overrides."""
