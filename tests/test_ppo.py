"""PPO integration tests (protocol of reference tests/test_ppo.py:36-71:
CartPole threshold > 35 after 5 epochs x 500 steps; Pendulum > -1300)."""
import numpy as np
import pytest
import torch
import torch.nn as nn

from rl_replicas_amd import envs
from rl_replicas_amd.algorithms import PPO
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.networks import MLP
from rl_replicas_amd.policies import CategoricalPolicy, GaussianPolicy
from rl_replicas_amd.samplers import BatchSampler, VectorSampler
from rl_replicas_amd.utils import set_seed_for_libraries
from rl_replicas_amd.value_function import ValueFunction


def make_ppo_cartpole(env, seed=0, sampler=None):
    obs_dim = env.observation_space.shape[0]
    n_act = env.action_space.n
    pnet = MLP([obs_dim, 64, 32, n_act])
    policy = CategoricalPolicy(pnet, torch.optim.Adam(pnet.parameters(), lr=3e-4))
    vnet = MLP([obs_dim, 64, 32, 1])
    vf = ValueFunction(vnet, torch.optim.Adam(vnet.parameters(), lr=1e-3))
    return PPO(policy, vf, env, sampler or BatchSampler(env, seed=seed))


def test_ppo_cartpole_learns(tmp_path):
    env = envs.make("CartPole-v1")
    model = make_ppo_cartpole(env)
    model.learn(num_epochs=5, batch_size=500, output_dir=str(tmp_path))
    returns, _ = Evaluator(seed=0).evaluate(model.policy, envs.make("CartPole-v1"), 3)
    assert np.mean(returns) > 35.0


def test_ppo_pendulum_learns(tmp_path):
    env = envs.make("Pendulum-v1")
    obs_dim = env.observation_space.shape[0]
    act_dim = env.action_space.shape[0]
    pnet = MLP([obs_dim, 64, 32, act_dim])
    log_std = nn.Parameter(-0.5 * torch.ones(act_dim))
    policy = GaussianPolicy(
        pnet, torch.optim.Adam(list(pnet.parameters()) + [log_std], lr=3e-4), log_std
    )
    vnet = MLP([obs_dim, 64, 32, 1])
    vf = ValueFunction(vnet, torch.optim.Adam(vnet.parameters(), lr=1e-3))
    model = PPO(policy, vf, env, BatchSampler(env, seed=0))
    model.learn(num_epochs=5, batch_size=500, output_dir=str(tmp_path))
    returns, _ = Evaluator(seed=0).evaluate(model.policy, envs.make("Pendulum-v1"), 3)
    assert np.mean(returns) > -1400.0


def test_ppo_with_vector_sampler_learns(tmp_path):
    """The MI355X throughput sampler trains PPO just as well."""
    env = envs.make("CartPole-v1")
    venv = envs.VectorEnv("CartPole-v1", num_envs=10)
    model = make_ppo_cartpole(env, sampler=VectorSampler(venv, seed=0))
    model.learn(num_epochs=10, batch_size=500, output_dir=str(tmp_path))
    returns, _ = Evaluator(seed=0).evaluate(model.policy, envs.make("CartPole-v1"), 3)
    assert np.mean(returns) > 35.0


def test_ppo_deterministic_across_runs(tmp_path):
    def run(d, seed=11):
        set_seed_for_libraries(seed)
        env = envs.make("CartPole-v1")
        model = make_ppo_cartpole(env, seed=seed)
        model.learn(num_epochs=2, batch_size=300, output_dir=str(d))
        returns, _ = Evaluator(seed=seed).evaluate(model.policy, envs.make("CartPole-v1"), 3)
        return returns

    assert run(tmp_path / "a") == run(tmp_path / "b")
