"""Checkpoint schema + resume tests.

The reference emits a fixed dict schema with nn.Sequential tensor
naming (`network.{0,2,4}.{weight,bias}`) and has NO load path
(SURVEY.md §5.4); we verify the exact schema and the added
load_model() roundtrip."""
import numpy as np
import torch
import torch.nn as nn

from rl_replicas_amd import envs
from rl_replicas_amd.algorithms import PPO, TD3
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.networks import MLP
from rl_replicas_amd.policies import CategoricalPolicy, DeterministicPolicy, RandomPolicy
from rl_replicas_amd.q_function import QFunction
from rl_replicas_amd.replay_buffer import ReplayBuffer
from rl_replicas_amd.samplers import BatchSampler


def make_ppo(env):
    pnet = MLP([4, 64, 32, 2])
    policy = CategoricalPolicy(pnet, torch.optim.Adam(pnet.parameters(), lr=3e-4))
    vnet = MLP([4, 64, 32, 1])
    from rl_replicas_amd.value_function import ValueFunction

    vf = ValueFunction(vnet, torch.optim.Adam(vnet.parameters(), lr=1e-3))
    return PPO(policy, vf, env, BatchSampler(env, seed=0))


def test_on_policy_checkpoint_schema(tmp_path):
    env = envs.make("CartPole-v1")
    model = make_ppo(env)
    model.current_total_steps = 123
    path = str(tmp_path / "model.pt")
    model.save_model(7, path)

    ckpt = torch.load(path, weights_only=False)
    assert set(ckpt.keys()) == {
        "epoch",
        "total_steps",
        "policy_state_dict",
        "policy_optimizer_state_dict",
        "value_function_state_dict",
        "value_function_optimizer_state_dict",
    }
    assert ckpt["epoch"] == 7 and ckpt["total_steps"] == 123
    # exact nn.Sequential naming: network.{0,2,4}.{weight,bias}
    assert set(ckpt["policy_state_dict"].keys()) == {
        "network.0.weight",
        "network.0.bias",
        "network.2.weight",
        "network.2.bias",
        "network.4.weight",
        "network.4.bias",
    }
    assert ckpt["policy_state_dict"]["network.0.weight"].shape == (64, 4)


def test_td3_checkpoint_schema(tmp_path):
    env = envs.make("Pendulum-v1")
    pnet = MLP([3, 32, 1], activation_function=nn.ReLU, output_activation_function=nn.Tanh)
    policy = DeterministicPolicy(pnet, torch.optim.Adam(pnet.parameters(), 1e-3))
    q1n = MLP([4, 32, 1], activation_function=nn.ReLU)
    q2n = MLP([4, 32, 1], activation_function=nn.ReLU)
    model = TD3(
        policy,
        RandomPolicy(env.action_space),
        QFunction(q1n, torch.optim.Adam(q1n.parameters(), 1e-3)),
        QFunction(q2n, torch.optim.Adam(q2n.parameters(), 1e-3)),
        env,
        BatchSampler(env, seed=0, is_continuous=True),
        ReplayBuffer(1000),
        Evaluator(seed=1),
    )
    model.current_total_steps = 50
    path = str(tmp_path / "model.pt")
    model.save_model(3, path)
    ckpt = torch.load(path, weights_only=False)
    assert set(ckpt.keys()) == {
        "epoch",
        "total_steps",
        "policy_state_dict",
        "policy_optimizer_state_dict",
        "target_policy_state_dict",
        "q_function_1_state_dict",
        "q_function_1_optimizer_state_dict",
        "target_q_function_1_state_dict",
        "q_function_2_state_dict",
        "q_function_2_optimizer_state_dict",
        "target_q_function_2_state_dict",
    }


def test_load_model_roundtrip(tmp_path):
    env = envs.make("CartPole-v1")
    model = make_ppo(env)
    model.learn(num_epochs=1, batch_size=200, model_saving_interval=200, output_dir=str(tmp_path))
    path = str(tmp_path / "model.pt")

    model2 = make_ppo(env)
    epoch = model2.load_model(path)
    assert epoch == 1
    assert model2.current_total_steps == 200
    for k, v in model.policy.network.state_dict().items():
        torch.testing.assert_close(model2.policy.network.state_dict()[k], v)
    # optimizer state restored too (Adam moments exist after training)
    st = model2.policy.optimizer.state_dict()["state"]
    assert len(st) > 0 and "exp_avg" in next(iter(st.values()))


def test_reference_checkpoint_loads():
    """A checkpoint written by the reference library (same key schema,
    same tensor naming) loads into our networks."""
    # simulate a reference checkpoint dict
    ref_policy_sd = MLP([4, 64, 32, 2]).state_dict()  # network.N.* naming
    env = envs.make("CartPole-v1")
    model = make_ppo(env)
    model.policy.network.load_state_dict(ref_policy_sd)
    for k in ref_policy_sd:
        torch.testing.assert_close(model.policy.network.state_dict()[k], ref_policy_sd[k])


def test_rl_replicas_drop_in_alias():
    """Reference-style imports work through the rl_replicas shim."""
    from rl_replicas.algorithms import PPO as AliasPPO
    from rl_replicas_amd.algorithms import PPO

    assert AliasPPO is PPO
    from rl_replicas.utils import discounted_cumulative_sums
    import numpy as np

    np.testing.assert_allclose(
        discounted_cumulative_sums(np.array([1.0, 1.0]), 0.5), [1.5, 1.0]
    )
