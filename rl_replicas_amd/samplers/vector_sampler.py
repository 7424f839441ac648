"""Vectorized rollout sampler — the MI355X throughput path.

The reference alternates one policy inference with one `env.step` per
sample (batch_sampler.py:55-99), capping the whole node at one serial
env.  Here N batched env instances advance together: each sampler step
is ONE batched policy forward over obs[N, O] (one H2D + one fused MLP
kernel on GPU) plus ONE vectorized env transition, so `num_samples`
total steps take `num_samples / N` policy calls.

Episode-slicing semantics match BatchSampler exactly per instance:
episodes end on terminated|truncated (both flagged `done`, matching
the reference's bootstrap semantics) or at epoch end (trajectory cut,
`done=False`, bootstrapped with V(s_last)); `is_continuous=True`
retains env state across calls; the env is seeded on first reset only.

Episodes are emitted in instance-major order (all of instance 0's
slices, then instance 1's, ...) — deterministic given the seed.
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np

from rl_replicas_amd.envs.vector import VectorEnv
from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.samplers.batch_sampler import _EpisodeAccumulator
from rl_replicas_amd.samplers.sampler import Sampler


class VectorSampler(Sampler):
    def __init__(self, vector_env: VectorEnv, seed: Optional[int] = None, is_continuous: bool = False):
        self.env = vector_env
        self.seed = seed
        self.is_continuous = is_continuous
        self.observations: Optional[np.ndarray] = None
        self.num_envs = vector_env.num_envs

    def sample(self, num_samples: int, policy: Policy) -> Experience:
        n_envs = self.num_envs
        if num_samples % n_envs != 0:
            raise ValueError(
                f"num_samples ({num_samples}) must be divisible by num_envs ({n_envs})"
            )
        steps = num_samples // n_envs

        if self.observations is None:
            self.observations = self.env.reset(seed=self.seed)
        elif not self.is_continuous:
            self.observations = self.env.reset()

        accs = [_EpisodeAccumulator() for _ in range(n_envs)]
        # per-instance episode slices, stitched instance-major at the end
        slices: List[List] = [[] for _ in range(n_envs)]

        for step in range(steps):
            obs = self.observations
            actions = np.asarray(policy.get_action_numpy(obs))
            next_obs, rewards, terminated, truncated, final_obs = self.env.step(actions)
            done = terminated | truncated
            epoch_ended = step == steps - 1

            for i in range(n_envs):
                accs[i].push(obs[i], actions[i], rewards[i], bool(done[i]))
                if done[i] or epoch_ended:
                    # final_obs holds the true successor state (pre-autoreset)
                    slices[i].append((accs[i], np.asarray(final_obs[i])))
                    accs[i] = _EpisodeAccumulator()
            self.observations = next_obs

        experience = Experience()
        for i in range(n_envs):
            for acc, last_obs in slices[i]:
                acc.flush_into(experience, last_obs)
        return experience
