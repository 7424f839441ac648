"""Shared model-building helpers for the benchmark scripts.

Network shapes and hyperparameters reproduce the reference's published
configs exactly (BASELINE.md "Config behind the numbers"; reference
benchmarks/run_vpg.py:28-42, run_td3.py):
  on-policy:  Gaussian policy MLP [obs, 64, 32, act] tanh, log_std -0.5,
              Adam 3e-4 (TRPO: CG optimizer defaults), value MLP
              [obs, 64, 32, 1] Adam 1e-3, 750 epochs x 4000 steps.
  off-policy: policy MLP [obs, 256, 256, act] ReLU + Tanh head, Q nets
              [obs+act, 256, 256, 1] ReLU, Adam 1e-3, 20000 x 50 steps.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from rl_replicas_amd import envs, ops
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.networks import MLP
from rl_replicas_amd.policies import (
    CategoricalPolicy,
    DeterministicPolicy,
    GaussianPolicy,
    RandomPolicy,
)
from rl_replicas_amd.q_function import QFunction
from rl_replicas_amd.replay_buffer import ReplayBuffer
from rl_replicas_amd.samplers import BatchSampler, VectorSampler
from rl_replicas_amd.utils import set_seed_for_libraries
from rl_replicas_amd.value_function import ValueFunction


def pick_device(device: Optional[str]) -> str:
    if device is not None:
        return device
    return "cuda" if torch.cuda.is_available() else "cpu"


def make_sampler(env_id: str, seed: int, num_envs: int, is_continuous: bool = False,
                 env_mode: str = "cpu", device: str = "cpu"):
    if env_mode == "device":
        # GPU-resident rollout (envs/device.py); synthetic MuJoCo shapes
        # and Pendulum have lockstep device implementations
        from rl_replicas_amd.samplers import DeviceSampler

        if env_id == "Pendulum-v1":
            denv = envs.DevicePendulumEnv(num_envs=num_envs, device=device)
        else:
            denv = envs.DeviceVectorEnv(env_id, num_envs=num_envs, device=device)
        return denv, DeviceSampler(denv, seed=seed, is_continuous=is_continuous)
    if num_envs > 1:
        venv = envs.VectorEnv(env_id, num_envs=num_envs)
        return venv, VectorSampler(venv, seed=seed, is_continuous=is_continuous)
    env = envs.make(env_id)
    return env, BatchSampler(env, seed=seed, is_continuous=is_continuous)


def build_on_policy(env_id: str, seed: int, device: Optional[str], num_envs: int,
                    optimizer: str, env_mode: str = "cpu"):
    set_seed_for_libraries(seed)
    dev = pick_device(device)
    env, sampler = make_sampler(env_id, seed, num_envs, env_mode=env_mode, device=dev)
    obs_dim = env.observation_space.shape[0]

    if hasattr(env.action_space, "n"):
        net = MLP([obs_dim, 64, 32, env.action_space.n]).to(dev)
        if optimizer == "cg":
            from rl_replicas_amd.optimizers import ConjugateGradientOptimizer

            policy = CategoricalPolicy(net, ConjugateGradientOptimizer(net.parameters()))
        else:
            policy = CategoricalPolicy(net, ops.make_adam(net.parameters(), lr=3e-4))
    else:
        act_dim = env.action_space.shape[0]
        net = MLP([obs_dim, 64, 32, act_dim]).to(dev)
        log_std = nn.Parameter(-0.5 * torch.ones(act_dim, device=dev))
        params = list(net.parameters()) + [log_std]
        if optimizer == "cg":
            from rl_replicas_amd.optimizers import ConjugateGradientOptimizer

            policy = GaussianPolicy(net, ConjugateGradientOptimizer(params), log_std)
        else:
            policy = GaussianPolicy(net, ops.make_adam(params, lr=3e-4), log_std)

    vnet = MLP([obs_dim, 64, 32, 1]).to(dev)
    value_function = ValueFunction(vnet, ops.make_adam(vnet.parameters(), lr=1e-3))
    return env, sampler, policy, value_function


def build_off_policy(env_id: str, seed: int, device: Optional[str], num_envs: int,
                     twin: bool, env_mode: str = "cpu"):
    set_seed_for_libraries(seed)
    dev = pick_device(device)
    env, sampler = make_sampler(env_id, seed, num_envs, is_continuous=True,
                                env_mode=env_mode, device=dev)
    obs_dim = env.observation_space.shape[0]
    act_dim = env.action_space.shape[0]

    pnet = MLP([obs_dim, 256, 256, act_dim], activation_function=nn.ReLU, output_activation_function=nn.Tanh).to(dev)
    policy = DeterministicPolicy(pnet, ops.make_adam(pnet.parameters(), lr=1e-3))
    exploration = RandomPolicy(env.action_space)

    def make_q():
        qnet = MLP([obs_dim + act_dim, 256, 256, 1], activation_function=nn.ReLU).to(dev)
        return QFunction(qnet, ops.make_adam(qnet.parameters(), lr=1e-3))

    qs = [make_q() for _ in range(2 if twin else 1)]
    buffer = ReplayBuffer(int(1e6), device=dev if dev.startswith("cuda") else None)
    evaluator = Evaluator(seed=seed)
    return env, sampler, policy, exploration, qs, buffer, evaluator
