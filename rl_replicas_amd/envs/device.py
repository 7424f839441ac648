"""GPU-resident vectorized synthetic environments.

The CPU `VectorEnv` path steps numpy dynamics on the host, which makes
every rollout step pay a D2H (actions) + H2D (observations) round trip
plus host arithmetic.  For the synthetic MuJoCo-shaped benchmark envs
(`envs/synthetic.py`) the dynamics are a few small matmuls — exactly
the kind of work the GPU does for free next to the policy forward — so
this module keeps the *entire* environment state in HBM (Isaac-Gym
style): policy forward, Philox action sampling, env transition and
reward all run on device and the epoch's rollout never touches the
host.

Semantics match `SyntheticEnv`/`VectorEnv` (same dynamics matrices —
they are taken from the registered numpy env — same tanh/clip/reward
formulas, same 1000-step horizon with autoreset); only the RNG streams
differ (torch Philox here vs numpy PCG64 there), which is the same
freedom the reference has across gymnasium versions.  Episodes advance
in lockstep: every instance resets together, so truncation boundaries
are host-known arithmetic and episode slicing costs nothing.

No reference counterpart (the reference steps ONE gymnasium env on the
CPU, batch_sampler.py:55-99); this is MI355X-first design, validated
against the numpy env in tests/test_device_env.py.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from .core import make
from .synthetic import SyntheticEnv


class DeviceVectorEnv:
    """N synthetic env instances as device tensors, stepped in lockstep.

    s' = tanh(s A + clip(a) B + noise*eps);  r = <s', w> - 0.1|a|^2
    (identical to SyntheticEnv._step_b, synthetic.py:72-78).
    """

    def __init__(self, env: str, num_envs: int, device="cpu", **make_kwargs):
        base = make(env, **make_kwargs)
        if not isinstance(base, SyntheticEnv):
            raise TypeError(
                f"DeviceVectorEnv supports the synthetic benchmark envs (got {type(base).__name__})"
            )
        self.num_envs = int(num_envs)
        self.device = torch.device(device)
        self.observation_space = base.observation_space
        self.action_space = base.action_space
        self.spec = base.spec
        self.noise = float(base.noise)
        # identical dynamics to the numpy env (same seed-derived matrices)
        self.A = torch.from_numpy(base.A).to(self.device)
        self.B = torch.from_numpy(base.B).to(self.device)
        self.w = torch.from_numpy(base.w).to(self.device)
        self._gen = torch.Generator(device=self.device)
        self._gen.manual_seed(torch.initial_seed() & 0x7FFFFFFFFFFFFFFF)
        self.state: Optional[Tensor] = None
        self._elapsed = 0  # lockstep: one counter for all instances
        # HIP fast path RNG: Philox keyed (seed ^ salt, offset).  The salt
        # decorrelates the env stream from the policy's action-sampling
        # stream, which is keyed on the same torch seed.
        self._philox_seed = (torch.initial_seed() ^ 0x9E3779B97F4A7C15) & 0x7FFFFFFFFFFFFFFF
        self._philox_offset = 0

    def seed(self, seed: Optional[int]) -> None:
        if seed is not None:
            self._gen.manual_seed(int(seed))
            self._philox_seed = (int(seed) ^ 0x9E3779B97F4A7C15) & 0x7FFFFFFFFFFFFFFF
            self._philox_offset = 0
            self.action_space.seed(seed + 1000)  # VectorEnv.seed contract

    def _wants_hip(self) -> bool:
        from rl_replicas_amd import ops

        return self.device.type == "cuda" and ops.wants_hip(
            self.state if self.state is not None else self.A
        )

    def _init_state(self) -> Tensor:
        return 0.1 * torch.randn(
            self.num_envs, self.A.shape[0], generator=self._gen, device=self.device
        )

    def reset(self, *, seed: Optional[int] = None) -> Tensor:
        if seed is not None:
            self.seed(seed)
        if self._wants_hip():
            from rl_replicas_amd import ops

            ext = ops._load_extension()
            self.state = ext.synthetic_env_reset(
                self.num_envs, self.A.shape[0], self.A,
                self._philox_seed, self._philox_offset,
            )
            self._philox_offset += 1
        else:
            self.state = self._init_state()
        self._elapsed = 0
        return self.state

    def step(self, actions: Tensor) -> Tuple[Tensor, Tensor, bool, Tensor]:
        """-> (obs [N,O], reward [N], truncated: host bool, final_obs [N,O]).

        `truncated` is a scalar (lockstep horizon; synthetic envs never
        terminate early).  On truncation all instances autoreset; `obs`
        is post-reset, `final_obs` the true successor (GAE bootstrap).
        """
        self._elapsed += 1
        truncated = (
            self.spec.max_episode_steps is not None
            and self._elapsed >= self.spec.max_episode_steps
        )
        if self._wants_hip():
            from rl_replicas_amd import ops

            ext = ops._load_extension()
            # ONE kernel: clip + sA + aB + Philox noise + tanh + reward
            # (+ the autoreset init states on lockstep horizon boundaries)
            state, reward, final_obs = ext.synthetic_env_step(
                self.state, actions.contiguous(), self.A, self.B, self.w,
                self.noise, self._philox_seed, self._philox_offset, truncated,
            )
            self._philox_offset += 2
        else:
            a = actions.clamp(-1.0, 1.0)
            eps = torch.randn(
                self.num_envs, self.A.shape[0], generator=self._gen, device=self.device
            )
            state = torch.tanh(
                torch.addmm(self.noise * eps, self.state, self.A).addmm_(a, self.B)
            )
            reward = state @ self.w - 0.1 * (a * a).sum(dim=1)
            final_obs = state
            if truncated:
                state = self._init_state()
        if truncated:
            self._elapsed = 0
        self.state = state
        return state, reward, truncated, final_obs

    def close(self) -> None:
        pass


class DevicePendulumEnv:
    """Pendulum-v1 as a lockstep device vector env (pure torch ops — the
    identical code runs on CPU and GPU; no HIP kernel needed at N x 2
    state sizes).  Dynamics/reward formulas match `classic.PendulumEnv`
    exactly (fp64 state, fp32 obs); only the init RNG stream differs
    (torch generator vs numpy).  Pendulum never terminates early, so the
    lockstep-truncation contract of `DeviceSampler` holds.
    """

    def __init__(self, num_envs: int, device="cpu"):
        from .classic import PendulumEnv

        base = PendulumEnv()
        self.num_envs = int(num_envs)
        self.device = torch.device(device)
        self.observation_space = base.observation_space
        self.action_space = base.action_space
        self.spec = base.spec
        self._gen = torch.Generator(device=self.device)
        self._gen.manual_seed(torch.initial_seed() & 0x7FFFFFFFFFFFFFFF)
        self.state: Optional[Tensor] = None  # [N, 2] float64: theta, theta_dot
        self._elapsed = 0

    def seed(self, seed: Optional[int]) -> None:
        if seed is not None:
            self._gen.manual_seed(int(seed))
            self.action_space.seed(seed + 1000)

    def _init_state(self) -> Tensor:
        u = torch.rand(self.num_envs, 2, generator=self._gen, device=self.device,
                       dtype=torch.float64)
        import math

        return (2.0 * u - 1.0) * torch.tensor([math.pi, 1.0], device=self.device,
                                              dtype=torch.float64)

    def _obs(self) -> Tensor:
        th, thdot = self.state[:, 0], self.state[:, 1]
        return torch.stack([torch.cos(th), torch.sin(th), thdot], dim=1).float()

    def reset(self, *, seed: Optional[int] = None) -> Tensor:
        if seed is not None:
            self.seed(seed)
        self.state = self._init_state()
        self._elapsed = 0
        return self._obs()

    def step(self, actions: Tensor) -> Tuple[Tensor, Tensor, bool, Tensor]:
        import math

        th, thdot = self.state[:, 0], self.state[:, 1]
        u = actions.to(torch.float64).reshape(self.num_envs, -1)[:, 0].clamp(-2.0, 2.0)
        ang = torch.remainder(th + math.pi, 2 * math.pi) - math.pi
        costs = ang * ang + 0.1 * thdot * thdot + 0.001 * u * u
        newthdot = (thdot + (15.0 * torch.sin(th) + 3.0 * u) * 0.05).clamp(-8.0, 8.0)
        newth = th + newthdot * 0.05
        self.state = torch.stack([newth, newthdot], dim=1)
        self._elapsed += 1
        truncated = (
            self.spec.max_episode_steps is not None
            and self._elapsed >= self.spec.max_episode_steps
        )
        final_obs = self._obs()
        obs = final_obs
        if truncated:
            self.state = self._init_state()
            self._elapsed = 0
            obs = self._obs()
        reward = (-costs).float()
        return obs, reward, truncated, final_obs

    def close(self) -> None:
        pass
