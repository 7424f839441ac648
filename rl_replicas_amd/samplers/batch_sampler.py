"""Serial rollout sampler.

Behavioral parity with the reference `BatchSampler` (reference:
src/rl_replicas/samplers/batch_sampler.py:14-101): steps ONE env
`num_samples` times, slicing the stream into episodes on done or
epoch-end; `is_continuous=True` retains the current observation across
`sample()` calls (off-policy mode), otherwise the env is reset at the
start of every call; the env is seeded only on the very first reset.

This is the compatibility path for user-provided serial envs.  The
throughput path that feeds the GPU learner is `VectorSampler`
(vector_sampler.py), which preserves these episode-slicing semantics
over N batched env instances.
"""
from __future__ import annotations

import logging
from typing import List, Optional

import numpy as np

from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.samplers.sampler import Sampler

logger = logging.getLogger(__name__)


class _EpisodeAccumulator:
    """Per-episode step buffers + the slice-and-flush bookkeeping."""

    def __init__(self) -> None:
        self.reset()

    def reset(self) -> None:
        self.observations: List[np.ndarray] = []
        self.actions: List[np.ndarray] = []
        self.rewards: List[float] = []
        self.dones: List[bool] = []
        self.episode_return = 0.0

    def push(self, obs: np.ndarray, action: np.ndarray, reward: float, done: bool) -> None:
        self.observations.append(obs)
        self.actions.append(action)
        self.rewards.append(float(reward))
        self.dones.append(bool(done))
        self.episode_return += float(reward)

    def flush_into(self, experience: Experience, last_observation: np.ndarray) -> None:
        experience.observations.append(self.observations)
        experience.actions.append(self.actions)
        experience.rewards.append(self.rewards)
        experience.dones.append(self.dones)
        experience.last_observations.append(last_observation)
        experience.episode_returns.append(self.episode_return)
        experience.episode_lengths.append(len(self.rewards))
        self.reset()


class BatchSampler(Sampler):
    def __init__(self, env, seed: Optional[int] = None, is_continuous: bool = False):
        self.env = env
        self.seed = seed
        self.is_continuous = is_continuous
        self.observation: Optional[np.ndarray] = None

    def sample(self, num_samples: int, policy: Policy) -> Experience:
        experience = Experience()
        acc = _EpisodeAccumulator()

        if self.observation is None:
            # first call: seed the env exactly once
            self.observation, _ = self.env.reset(seed=self.seed)
        elif not self.is_continuous:
            self.observation, _ = self.env.reset()

        for step in range(num_samples):
            obs = self.observation
            action = policy.get_action_numpy(obs)
            self.observation, reward, terminated, truncated, _ = self.env.step(action)
            done = bool(terminated or truncated)
            acc.push(obs, action, reward, done)

            epoch_ended = step == num_samples - 1
            if done or epoch_ended:
                if epoch_ended and not done:
                    logger.debug(
                        "Trajectory cut off at %d steps at epoch end", len(acc.rewards)
                    )
                acc.flush_into(experience, self.observation)
                if done:
                    self.observation, _ = self.env.reset()

        return experience
