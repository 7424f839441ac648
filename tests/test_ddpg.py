"""DDPG integration tests (protocol of reference tests/test_ddpg.py:
short Pendulum training with warm-up exploration, then threshold-based
assertion — exact-return asserts are replaced with own-stack
determinism)."""
import numpy as np
import torch
import torch.nn as nn

from rl_replicas_amd import envs
from rl_replicas_amd.algorithms import DDPG
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.networks import MLP
from rl_replicas_amd.policies import DeterministicPolicy, RandomPolicy
from rl_replicas_amd.q_function import QFunction
from rl_replicas_amd.replay_buffer import ReplayBuffer
from rl_replicas_amd.samplers import BatchSampler
from rl_replicas_amd.utils import set_seed_for_libraries


def make_ddpg(env, seed=0):
    obs_dim = env.observation_space.shape[0]
    act_dim = env.action_space.shape[0]
    pnet = MLP([obs_dim, 64, 64, act_dim], activation_function=nn.ReLU, output_activation_function=nn.Tanh)
    policy = DeterministicPolicy(pnet, torch.optim.Adam(pnet.parameters(), lr=1e-3))
    qnet = MLP([obs_dim + act_dim, 64, 64, 1], activation_function=nn.ReLU)
    q = QFunction(qnet, torch.optim.Adam(qnet.parameters(), lr=1e-3))
    return DDPG(
        policy,
        RandomPolicy(env.action_space),
        q,
        env,
        BatchSampler(env, seed=seed, is_continuous=True),
        ReplayBuffer(int(1e5)),
        Evaluator(seed=seed + 1),
    )


def run(tmp_path, seed=0):
    set_seed_for_libraries(seed)
    env = envs.make("Pendulum-v1")
    model = make_ddpg(env, seed=seed)
    model.learn(
        num_epochs=60,
        batch_size=50,
        num_start_steps=1000,
        num_steps_before_update=1000,
        num_evaluation_episodes=2,
        evaluation_interval=1000,
        output_dir=str(tmp_path),
    )
    returns, _ = Evaluator(seed=seed).evaluate(model.policy, envs.make("Pendulum-v1"), 3)
    return float(np.mean(returns))


def test_ddpg_pendulum_runs_and_not_catastrophic(tmp_path):
    mean_return = run(tmp_path)
    # 3000 steps is too short to solve Pendulum; assert sane behavior
    # (worst case random ~= -1900, solved ~= -150)
    assert -1900.0 < mean_return <= 0.0


def test_ddpg_deterministic_across_runs(tmp_path):
    assert run(tmp_path / "a", seed=4) == run(tmp_path / "b", seed=4)
