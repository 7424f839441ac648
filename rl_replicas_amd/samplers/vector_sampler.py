"""Vectorized rollout sampler — the MI355X throughput path.

The reference alternates one policy inference with one `env.step` per
sample (batch_sampler.py:55-99), capping the whole node at one serial
env.  Here N batched env instances advance together: each sampler step
is ONE batched policy forward over obs[N, O] (one H2D + one fused MLP
kernel on GPU) plus ONE vectorized env transition, so `num_samples`
total steps take `num_samples / N` policy calls.

The per-step work is O(1) Python (append five array references);
rollout storage is step-major, and episode slicing happens once at the
end of `sample()` with numpy boundary searches — no per-instance Python
loop in the hot path.  Episodes are emitted in instance-major order
(all of instance 0's slices, then instance 1's, ...), deterministic
given the seed, with per-episode data exposed as contiguous array
views (compatible with the reference's List[List[...]] Experience
protocol).

Episode semantics match BatchSampler exactly per instance: episodes end
on terminated|truncated (both flagged `done`, matching the reference's
bootstrap semantics) or at epoch end (trajectory cut, `done=False`,
bootstrapped with V(s_last)); `is_continuous=True` retains env state
across calls; the env is seeded on first reset only.
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from rl_replicas_amd.envs.vector import VectorEnv
from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import Policy
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

        # step-major storage: the hot loop only appends references
        obs_steps = []
        act_steps = []
        rew_steps = []
        done_steps = []
        final_steps = []
        for _ in range(steps):
            obs = self.observations
            actions = np.asarray(policy.get_action_numpy(obs))
            next_obs, rewards, terminated, truncated, final_obs = self.env.step(actions)
            obs_steps.append(obs)
            act_steps.append(actions)
            rew_steps.append(rewards)
            done_steps.append(terminated | truncated)
            final_steps.append(final_obs)
            self.observations = next_obs

        # [T, B, ...] views for slicing
        obs_arr = np.stack(obs_steps)
        act_arr = np.stack(act_steps)
        rew_arr = np.stack(rew_steps)
        done_arr = np.stack(done_steps)

        experience = Experience()
        for i in range(n_envs):
            done_i = done_arr[:, i]
            boundaries = np.flatnonzero(done_i)
            start = 0
            for b in boundaries:
                self._emit(
                    experience,
                    obs_arr[start : b + 1, i],
                    act_arr[start : b + 1, i],
                    rew_arr[start : b + 1, i],
                    done_i[start : b + 1],
                    np.asarray(final_steps[b][i]),
                )
                start = b + 1
            if start < steps:
                # epoch-end trajectory cut: bootstrap obs is the live state
                self._emit(
                    experience,
                    obs_arr[start:steps, i],
                    act_arr[start:steps, i],
                    rew_arr[start:steps, i],
                    done_i[start:steps],
                    np.asarray(self.observations[i]),
                )

        # flat (instance-major, time-ordered) view in O(1) numpy ops —
        # exactly the order the per-episode emission above produces
        lengths = np.asarray(experience.episode_lengths, dtype=np.int64)
        offsets = np.zeros(len(lengths) + 1, dtype=np.int32)
        np.cumsum(lengths, out=offsets[1:])
        flat_obs = np.ascontiguousarray(obs_arr.transpose(1, 0, *range(2, obs_arr.ndim))).reshape(
            steps * n_envs, *obs_arr.shape[2:]
        )
        flat_act = np.ascontiguousarray(act_arr.transpose(1, 0, *range(2, act_arr.ndim))).reshape(
            steps * n_envs, *act_arr.shape[2:]
        )
        experience.set_flat_cache(
            {
                "observations": flat_obs.astype(np.float32, copy=False),
                "actions": flat_act,
                "rewards": np.ascontiguousarray(rew_arr.T).reshape(-1).astype(np.float32),
                "episode_offsets": offsets,
                "episode_dones": np.asarray(experience.episode_dones, dtype=bool),
                "last_observations": np.stack(experience.last_observations).astype(np.float32),
            }
        )
        return experience

    @staticmethod
    def _emit(experience: Experience, obs, acts, rews, dones, last_obs) -> None:
        # contiguous per-episode views satisfy the List[List[...]] protocol
        experience.observations.append(obs)
        experience.actions.append(acts)
        experience.rewards.append(rews)
        experience.dones.append(dones)
        experience.last_observations.append(last_obs)
        experience.episode_returns.append(float(np.sum(rews)))
        experience.episode_lengths.append(int(len(rews)))
