"""Device-resident rollout sampler — zero host traffic in the hot loop.

Pairs with `envs.DeviceVectorEnv`: observations, actions, rewards and
env state all live in HBM, the policy forward + Philox sample + env
transition chain runs back-to-back on the compute stream, and the flat
batch handed to the algorithms (`Experience.set_flat_cache`) is made of
device tensors — `OnPolicyAlgorithm._prepare_batch` skips its pinned
H2D upload entirely.  The only per-epoch D2H is one small readback of
per-episode returns/lengths for metrics (a few hundred floats).

On GPU with a Gaussian MLP policy the whole epoch rollout is captured
once into a hipGraph and replayed per epoch (`ops/fused_rollout.py`);
otherwise an eager loop issues the same kernels.  Both paths produce
identical episode structure; randomness streams differ (device-counter
Philox vs host-counter Philox/torch generator), each deterministic
under its seed.

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

from typing import Dict, List, Optional, Tuple

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
        self._bufs = None  # eager-path buffers, reused across epochs
        self._graphs: Dict[tuple, object] = {}
        self._graph_ctr: Optional[torch.Tensor] = None

    def _get_bufs(self, steps: int, obs_dim: int, act_dim: int, device):
        if self._bufs is None or self._bufs[0].shape[0] != steps:
            N = self.num_envs
            self._bufs = (
                torch.empty(steps, N, obs_dim, device=device),
                torch.empty(steps, N, act_dim, device=device),
                torch.empty(steps, N, device=device),
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
        obs_dim = int(env.observation_space.shape[0])
        act_dim = int(env.action_space.shape[0])

        first = self._obs is None
        reset_epoch = first or not self.is_continuous
        horizon = env.spec.max_episode_steps
        start_elapsed = 0 if reset_epoch else env._elapsed
        # lockstep truncation boundaries are pure host arithmetic
        cuts: Tuple[int, ...] = tuple(
            t for t in range(steps)
            if horizon is not None and (start_elapsed + t + 1) % horizon == 0
        )

        from rl_replicas_amd.ops import fused_rollout

        # when horizon % steps != 0 the truncation pattern cycles through
        # up to horizon/gcd values — cap the graph cache and serve unseen
        # patterns eagerly instead of capturing without bound
        graph_ok = fused_rollout.supported(policy, env) and (
            (steps, cuts, reset_epoch) in self._graphs or len(self._graphs) < 8
        )
        if graph_ok:
            if first and self.seed is not None:
                env.seed(self.seed)
            obs_buf, act_buf, rew_buf, finals = self._sample_graphed(
                policy, steps, cuts, reset_epoch
            )
        else:
            obs_buf, act_buf, rew_buf, finals = self._sample_eager(
                policy, steps, cuts, reset_epoch, obs_dim, act_dim
            )
        env._elapsed = steps - (cuts[-1] + 1) if cuts else start_elapsed + steps

        return self._build_experience(
            steps, cuts, obs_buf, act_buf, rew_buf, finals, obs_dim, act_dim
        )

    # ------------------------------------------------------------------
    def _sample_graphed(self, policy, steps: int, cuts: Tuple[int, ...],
                        reset_epoch: bool):
        from rl_replicas_amd.ops import fused_rollout

        env = self.env
        if self._graph_ctr is None:
            self._graph_ctr = fused_rollout.make_counter(env.device)
        key = (steps, cuts, reset_epoch)
        g = self._graphs.get(key)
        if g is None:
            g = fused_rollout.GraphedRollout(
                policy, env, steps, cuts, reset_epoch, self._graph_ctr
            )
            self._graphs[key] = g
        if not reset_epoch:
            # live state into the carry slot (no-op when the previous epoch
            # used this same graph; needed across pattern/mode switches)
            g.obs_full[steps].copy_(self._obs)
        g.replay()
        self._obs = g.obs_full[steps]
        env.state = self._obs
        return g.obs_full[:steps], g.act_buf, g.rew_buf, g.final_tensors()

    # ------------------------------------------------------------------
    def _sample_eager(self, policy, steps: int, cuts: Tuple[int, ...],
                      reset_epoch: bool, obs_dim: int, act_dim: int):
        env = self.env
        if self._obs is None:
            self._obs = env.reset(seed=self.seed)
        elif reset_epoch:
            self._obs = env.reset()
        obs_buf, act_buf, rew_buf = self._get_bufs(steps, obs_dim, act_dim, env.device)

        finals: List[torch.Tensor] = []
        obs = self._obs
        for t in range(steps):
            obs_buf[t].copy_(obs)
            actions = policy.get_action_tensor(obs).to(
                device=obs.device, dtype=torch.float32
            )
            act_buf[t].copy_(actions)
            obs, reward, truncated, final_obs = env.step(actions)
            rew_buf[t].copy_(reward)
            if truncated:
                finals.append(final_obs)
        assert len(finals) == len(cuts)
        self._obs = obs
        return obs_buf, act_buf, rew_buf, finals

    # ------------------------------------------------------------------
    def _build_experience(self, steps: int, cuts: Tuple[int, ...], obs_buf,
                          act_buf, rew_buf, finals, obs_dim: int,
                          act_dim: int) -> Experience:
        N = self.num_envs
        device = self.env.device

        seg_bounds: List[tuple] = []  # (start, end_exclusive, done)
        start = 0
        for c in cuts:
            seg_bounds.append((start, c + 1, True))
            start = c + 1
        if start < steps:
            seg_bounds.append((start, steps, False))
        n_segs = len(seg_bounds)

        # flat (instance-major) device views for the train pipeline
        flat_obs = torch.empty(N * steps, obs_dim, device=device)
        flat_act = torch.empty(N * steps, act_dim, device=device)
        flat_rew = torch.empty(N * steps, device=device)
        flat_obs.view(N, steps, obs_dim).copy_(obs_buf.transpose(0, 1))
        flat_act.view(N, steps, act_dim).copy_(act_buf.transpose(0, 1))
        rew_t = flat_rew.view(N, steps)
        rew_t.copy_(rew_buf.transpose(0, 1))

        # bootstrap observations per (instance, segment)
        lasts = [finals[j] if done else self._obs for j, (_, _, done) in enumerate(seg_bounds)]
        last_obs = torch.stack(lasts, dim=1).reshape(N * n_segs, obs_dim)

        seg_lengths = [e - s for s, e, _ in seg_bounds]
        offsets_np = np.zeros(N * n_segs + 1, dtype=np.int32)
        np.cumsum(np.tile(seg_lengths, N), out=offsets_np[1:])
        dones_pattern = [d for _, _, d in seg_bounds]

        # one tiny D2H: per-episode returns for the metrics layer
        seg_returns = torch.stack(
            [rew_t[:, s:e].sum(dim=1) for s, e, _ in seg_bounds], dim=1
        ).cpu().numpy()  # [N, n_segs]

        # per-step successors + done flags (the replay-buffer view; cut
        # steps get the TRUE pre-reset successor and done=1)
        flat_next = torch.empty(N * steps, obs_dim, device=device)
        nv = flat_next.view(N, steps, obs_dim)
        if steps > 1:
            nv[:, : steps - 1].copy_(obs_buf[1:].transpose(0, 1))
        nv[:, steps - 1].copy_(self._obs)
        step_dones = torch.zeros(N, steps, device=device)
        for j, c in enumerate(cuts):
            nv[:, c].copy_(finals[j])
            step_dones[:, c] = 1.0

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
                "episode_offsets": torch.as_tensor(offsets_np, device=device),
                "episode_dones": torch.as_tensor(
                    np.tile(dones_pattern, N), device=device
                ),
                "last_observations": last_obs,
                "next_observations": flat_next,
                "step_dones": step_dones.reshape(-1),
            }
        )
        return experience
