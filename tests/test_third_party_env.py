"""Third-party (gymnasium-API) environment compatibility.

A reference user brings real `gymnasium.Env` objects.  gymnasium is not
installed in this stack, so these tests drive the framework with a mock
class that implements gymnasium's exact public surface — NOT our `Env`
base class — and pin that:

* `BatchSampler` (the reference's sampler path, batch_sampler.py:49-99)
  consumes it unchanged,
* `envs.register()` lets `envs.make()` (and therefore the off-policy
  algorithms' evaluation-env construction, ddpg.py:66) produce it,
* `SerialVectorEnv` vectorizes N instances of it,
* full PPO and DDPG runs train over it.
"""
import numpy as np
import pytest
import torch
import torch.nn as nn

from rl_replicas_amd import envs
from rl_replicas_amd.samplers import BatchSampler


class MockGymEnv:
    """gymnasium.Env lookalike: a 1-D integrator, |obs|<=2 terminates."""

    class _Spec:
        id = "MockIntegrator-v0"
        max_episode_steps = 25

    def __init__(self):
        self.spec = self._Spec()
        self.observation_space = envs.Box(-np.inf, np.inf, shape=(2,), dtype=np.float32)
        self.action_space = envs.Box(-1.0, 1.0, shape=(1,), dtype=np.float32)
        self._rng = np.random.default_rng()
        self._steps = 0

    def reset(self, *, seed=None, options=None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
            self.action_space.seed(seed + 1000)
        self._steps = 0
        self._state = self._rng.standard_normal(2).astype(np.float32) * 0.1
        return self._state.copy(), {}

    def step(self, action):
        a = float(np.clip(np.asarray(action).reshape(-1)[0], -1, 1))
        x, v = self._state
        v = 0.9 * v + 0.1 * a
        x = x + v
        self._state = np.array([x, v], dtype=np.float32)
        self._steps += 1
        terminated = bool(abs(x) > 2.0)
        truncated = self._steps >= self.spec.max_episode_steps and not terminated
        reward = -abs(x)
        return self._state.copy(), reward, terminated, truncated, {}

    def close(self):
        pass


@pytest.fixture(autouse=True)
def _register_mock():
    envs.register("MockIntegrator-v0", lambda **kw: MockGymEnv())
    yield


def _gaussian_policy(obs_dim, act_dim):
    from rl_replicas_amd import ops
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.policies import GaussianPolicy

    net = MLP([obs_dim, 16, act_dim])
    log_std = nn.Parameter(-0.5 * torch.ones(act_dim))
    return GaussianPolicy(
        net, ops.make_adam(list(net.parameters()) + [log_std], lr=3e-4), log_std
    )


def test_batch_sampler_consumes_gymnasium_api():
    torch.manual_seed(0)
    env = MockGymEnv()
    sampler = BatchSampler(env, seed=3)
    exp = sampler.sample(100, _gaussian_policy(2, 1))
    assert sum(exp.episode_lengths) == 100
    assert len(exp.episode_returns) == len(exp.last_observations)
    # terminated/truncated episodes are <= the env's horizon
    assert max(exp.episode_lengths) <= 25


def test_serial_vector_env_over_gymnasium_api():
    venv = envs.SerialVectorEnv([MockGymEnv for _ in range(4)])
    obs = venv.reset(seed=0)
    assert obs.shape == (4, 2)
    o, r, te, tr, fin = venv.step(np.zeros((4, 1), dtype=np.float32))
    assert o.shape == (4, 2) and r.shape == (4,) and fin.shape == (4, 2)


def test_ppo_trains_on_third_party_env(tmp_path):
    from rl_replicas_amd import ops
    from rl_replicas_amd.algorithms import PPO
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.value_function import ValueFunction

    torch.manual_seed(1)
    env = MockGymEnv()
    policy = _gaussian_policy(2, 1)
    vnet = MLP([2, 16, 1])
    vf = ValueFunction(vnet, ops.make_adam(vnet.parameters(), lr=1e-3))
    model = PPO(policy, vf, env, BatchSampler(env, seed=5),
                num_policy_gradients=5, num_value_gradients=5)
    model.learn(num_epochs=3, batch_size=100, output_dir=str(tmp_path))
    assert model.current_total_steps == 300


def test_ddpg_trains_on_third_party_env(tmp_path):
    from rl_replicas_amd.algorithms import DDPG
    from rl_replicas_amd.evaluator import Evaluator
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.policies import DeterministicPolicy, RandomPolicy
    from rl_replicas_amd.q_function import QFunction
    from rl_replicas_amd.replay_buffer import ReplayBuffer

    torch.manual_seed(2)
    env = MockGymEnv()
    pnet = MLP([2, 16, 1], activation_function=nn.ReLU,
               output_activation_function=nn.Tanh)
    policy = DeterministicPolicy(pnet, torch.optim.Adam(pnet.parameters(), lr=1e-3))
    qnet = MLP([3, 16, 1], activation_function=nn.ReLU)
    q = QFunction(qnet, torch.optim.Adam(qnet.parameters(), lr=1e-3))
    # the evaluation env comes from envs.make(env.spec.id) -> the
    # registered factory above (the gym.make equivalent, ddpg.py:66)
    model = DDPG(
        policy, RandomPolicy(env.action_space), q, env,
        BatchSampler(env, seed=8, is_continuous=True),
        ReplayBuffer(5000), Evaluator(seed=9),
    )
    model.learn(
        num_epochs=6, batch_size=50, num_start_steps=100,
        num_steps_before_update=100, num_train_steps=5,
        num_evaluation_episodes=2, evaluation_interval=100,
        output_dir=str(tmp_path),
    )
    assert model.current_total_steps == 300
