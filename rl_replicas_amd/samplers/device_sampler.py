"""Device-resident rollout sampler — zero host traffic in the hot loop.

Pairs with `envs.DeviceVectorEnv`: observations, actions, rewards and
env state all live in HBM, the policy forward + Philox sample + env
transition chain runs back-to-back on the compute stream, and the flat
batch handed to the algorithms (`Experience.set_flat_cache`) is made of
device tensors — `OnPolicyAlgorithm._prepare_batch` skips its pinned
H2D upload entirely.  The only per-epoch D2H is one small readback of
per-episode returns/lengths for metrics (a few hundred floats).

Episode semantics are those of `VectorSampler` (instance-major episode
order, epoch-end trajectory cut with `done=False`, truncation cut with
`done=True`, `is_continuous` retains env state across calls), with the
simplification the lockstep env guarantees: every instance truncates at
the same step, so episode boundaries are host arithmetic, never data.

Reference counterpart: the serial `BatchSampler` loop
(batch_sampler.py:55-99); this is its MI355X-first replacement for the
synthetic benchmark envs.
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from rl_replicas_amd.envs.device import DeviceVectorEnv
from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.samplers.sampler import Sampler


class DeviceSampler(Sampler):
    def __init__(self, device_env: DeviceVectorEnv, seed: Optional[int] = None, is_continuous: bool = False):
        self.env = device_env
        self.seed = seed
        self.is_continuous = is_continuous
        self.num_envs = device_env.num_envs
        self._obs: Optional[torch.Tensor] = None
        self._bufs = None  # (obs_buf[T,N,O], act_buf[T,N,A], rew_buf[T,N]) reused across epochs

    def _get_bufs(self, steps: int, obs_dim: int, act_dim: int, device):
        if self._bufs is None or self._bufs[0].shape[0] != steps:
            N = self.num_envs
            self._bufs = (
                torch.empty(steps, N, obs_dim, device=device),
                torch.empty(steps, N, act_dim, device=device),
                torch.empty(steps, N, device=device),
                torch.empty(N * steps, obs_dim, device=device),  # flat obs (instance-major)
                torch.empty(N * steps, act_dim, device=device),
                torch.empty(N * steps, device=device),
            )
        return self._bufs

    def sample(self, num_samples: int, policy: Policy) -> Experience:
        env = self.env
        N = self.num_envs
        if num_samples % N != 0:
            raise ValueError(
                f"num_samples ({num_samples}) must be divisible by num_envs ({N})"
            )
        steps = num_samples // N

        if self._obs is None:
            self._obs = env.reset(seed=self.seed)
        elif not self.is_continuous:
            self._obs = env.reset()

        obs_dim = env.A.shape[0]
        act_dim = env.B.shape[0]
        obs_buf, act_buf, rew_buf, flat_obs, flat_act, flat_rew = self._get_bufs(
            steps, obs_dim, act_dim, env.device
        )

        cuts: List[int] = []  # step indices where all instances truncated
        cut_final: List[torch.Tensor] = []
        obs = self._obs
        for t in range(steps):
            obs_buf[t].copy_(obs)
            actions = policy.get_action_tensor(obs)
            act_buf[t].copy_(actions)
            obs, reward, truncated, final_obs = env.step(actions)
            rew_buf[t].copy_(reward)
            if truncated:
                cuts.append(t)
                cut_final.append(final_obs)
        self._obs = obs

        # ---- episode structure (host arithmetic: boundaries are lockstep)
        seg_bounds: List[tuple] = []  # (start, end_exclusive, done)
        start = 0
        for c in cuts:
            seg_bounds.append((start, c + 1, True))
            start = c + 1
        if start < steps:
            seg_bounds.append((start, steps, False))
        n_segs = len(seg_bounds)

        # ---- flat (instance-major) device views for the train pipeline
        flat_obs.view(N, steps, obs_dim).copy_(obs_buf.transpose(0, 1))
        flat_act.view(N, steps, act_dim).copy_(act_buf.transpose(0, 1))
        rew_t = flat_rew.view(N, steps)
        rew_t.copy_(rew_buf.transpose(0, 1))

        # bootstrap observations per (instance, segment)
        lasts = [cut_final[j] if done else self._obs for j, (_, _, done) in enumerate(seg_bounds)]
        last_obs = torch.stack(lasts, dim=1).reshape(N * n_segs, obs_dim)

        seg_lengths = [e - s for s, e, _ in seg_bounds]
        offsets_np = np.zeros(N * n_segs + 1, dtype=np.int32)
        np.cumsum(np.tile(seg_lengths, N), out=offsets_np[1:])
        dones_pattern = [d for _, _, d in seg_bounds]

        # one tiny D2H: per-episode returns for the metrics layer
        seg_returns = torch.stack(
            [rew_t[:, s:e].sum(dim=1) for s, e, _ in seg_bounds], dim=1
        ).cpu().numpy()  # [N, n_segs]

        experience = Experience()
        experience.episode_returns = [float(r) for r in seg_returns.reshape(-1)]
        experience.episode_lengths = seg_lengths * N
        experience.dones = [
            np.concatenate([np.zeros(L - 1, dtype=bool), [d]])
            for L, d in zip(seg_lengths, dones_pattern)
        ] * N
        experience.set_flat_cache(
            {
                "observations": flat_obs,
                "actions": flat_act,
                "rewards": flat_rew,
                "episode_offsets": torch.as_tensor(offsets_np, device=env.device),
                "episode_dones": torch.as_tensor(
                    np.tile(dones_pattern, N), device=env.device
                ),
                "last_observations": last_obs,
            }
        )
        return experience
