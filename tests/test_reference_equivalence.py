"""Update-equivalence harness: our CPU path vs the reference implementation.

Real MuJoCo is unavailable offline, so the reference's published return
curves can't be reproduced directly (round-1 VERDICT.md, gap #2).  This
suite proves the ALGORITHM MATH instead: the actual reference modules
(/root/reference/src, gymnasium/tensorboard stubbed — see
tests/_reference_loader.py) and our stack are fed byte-identical
Experience/minibatch data from identically initialized networks, and the
resulting losses, KL, GAE pipeline, CG step and post-update parameters
must agree to fp32 tolerance for all five algorithms:

  VPG/PPO/TRPO  — full train(experience) epoch (reference vpg.py:127-192,
                  ppo.py:139-223, trpo.py:130-240)
  DDPG/TD3      — full train(buffer, n, mb) loop under matched RNG
                  streams (reference ddpg.py:195-253, td3.py:214-263)
  CG optimizer  — step() on identical closures (reference CGO:59-98)
  utils         — scan/GAE/bootstrap/normalize/polyak primitives

Our HIP kernels are separately pinned to this CPU path by the GPU
numerics tests, so equivalence is transitive: kernels == our CPU path ==
reference.
"""
from __future__ import annotations

import numpy as np
import pytest
import torch
import torch.nn as nn

from tests._reference_loader import (
    FakeMetricsManager,
    load_reference,
    make_fake_gym_env,
)

OBS_DIM, ACT_DIM = 3, 2


# ---------------------------------------------------------------------------
# fixtures: identical rollout data + identically initialized networks
# ---------------------------------------------------------------------------
def _make_rollout(rng: np.random.Generator, n_episodes=4, obs_dim=OBS_DIM, act_dim=ACT_DIM, lengths=None):
    """Ragged synthetic episodes; last episode truncated (not done)."""
    observations, actions, rewards, dones = [], [], [], []
    last_observations, episode_returns, episode_lengths = [], [], []
    lengths = lengths or [7, 12, 5, 9]
    for e in range(n_episodes):
        L = lengths[e % len(lengths)]
        obs = [rng.normal(size=obs_dim).astype(np.float32) for _ in range(L)]
        act = [rng.normal(size=act_dim).astype(np.float32) for _ in range(L)]
        rew = [float(rng.normal()) for _ in range(L)]
        done = [False] * (L - 1) + [e != n_episodes - 1]  # last episode truncated
        observations.append(obs)
        actions.append(act)
        rewards.append(rew)
        dones.append(done)
        last_observations.append(rng.normal(size=obs_dim).astype(np.float32))
        episode_returns.append(float(np.sum(rew)))
        episode_lengths.append(L)
    return dict(
        observations=observations,
        actions=actions,
        rewards=rewards,
        dones=dones,
        last_observations=last_observations,
        episode_returns=episode_returns,
        episode_lengths=episode_lengths,
    )


def _experience_pair(rollout):
    from rl_replicas_amd.experience import Experience as OurExperience

    ref = load_reference()
    import copy

    return (
        ref["Experience"](**copy.deepcopy(rollout)),
        OurExperience(**copy.deepcopy(rollout)),
    )


def _matched_mlps(sizes, seed=7, **mlp_kwargs):
    """(reference MLP, our MLP) with identical weights."""
    from rl_replicas_amd.networks import MLP as OurMLP

    ref = load_reference()
    torch.manual_seed(seed)
    ours = OurMLP(sizes, **mlp_kwargs)
    theirs = ref["MLP"](sizes=sizes, **mlp_kwargs)
    theirs.load_state_dict(ours.state_dict())  # same network.N.* naming contract
    return theirs, ours


def _assert_modules_close(ref_module, our_module, rtol=1e-4, atol=1e-6, label=""):
    ref_sd = ref_module.state_dict()
    our_sd = our_module.state_dict()
    assert ref_sd.keys() == our_sd.keys(), label
    for k in ref_sd:
        torch.testing.assert_close(
            our_sd[k], ref_sd[k], rtol=rtol, atol=atol, msg=f"{label}:{k}"
        )


# ---------------------------------------------------------------------------
# utils primitives
# ---------------------------------------------------------------------------
def test_utils_equivalence():
    ref = load_reference()
    from rl_replicas_amd import utils as ours

    rng = np.random.default_rng(0)
    x = rng.normal(size=37).astype(np.float32)

    np.testing.assert_allclose(
        ours.discounted_cumulative_sums(x, 0.99),
        ref["utils"].discounted_cumulative_sums(x, 0.99),
        rtol=1e-6,
    )

    values = rng.normal(size=38).astype(np.float32)  # rewards have the bootstrap appended
    xr = np.append(x, 0.0).astype(np.float32)
    np.testing.assert_allclose(
        ours.gae(xr, 0.99, values, 0.97),
        ref["utils"].gae(xr, 0.99, values, 0.97),
        rtol=1e-5,
        atol=1e-6,
    )

    rewards = [[1.0, 2.0, 3.0], [0.5, -1.0]]
    dones = [True, False]
    last_values = [10.0, 20.0]
    ref_boot = ref["utils"].bootstrap_rewards_with_last_values(rewards, dones, last_values)
    our_boot = ours.bootstrap_rewards_with_last_values(rewards, dones, last_values)
    for a, b in zip(ref_boot, our_boot):
        np.testing.assert_allclose(np.asarray(a), np.asarray(b))

    t = torch.from_numpy(rng.normal(size=64).astype(np.float32))
    torch.testing.assert_close(
        ours.normalize_tensor(t.clone()), ref["utils"].normalize_tensor(t.clone())
    )

    torch.manual_seed(3)
    p_ref = [torch.randn(4, 4), torch.randn(4)]
    t_ref = [torch.randn(4, 4), torch.randn(4)]
    p_our = [a.clone() for a in p_ref]
    t_our = [a.clone() for a in t_ref]
    ref["utils"].polyak_average(p_ref, t_ref, 0.995)
    ours.polyak_average(p_our, t_our, 0.995)
    for a, b in zip(t_ref, t_our):
        torch.testing.assert_close(b, a)


# ---------------------------------------------------------------------------
# on-policy: full train(experience) equivalence
# ---------------------------------------------------------------------------
def _build_onpolicy_pair(algo_name: str):
    ref = load_reference()
    from rl_replicas_amd.algorithms import PPO as OurPPO
    from rl_replicas_amd.algorithms import TRPO as OurTRPO
    from rl_replicas_amd.algorithms import VPG as OurVPG
    from rl_replicas_amd.policies import GaussianPolicy as OurGaussianPolicy
    from rl_replicas_amd.value_function import ValueFunction as OurValueFunction

    fake_env = make_fake_gym_env("Fake-v0", OBS_DIM, ACT_DIM, 1.0)

    pnet_ref, pnet_our = _matched_mlps([OBS_DIM, 8, 8, ACT_DIM], seed=11)
    vnet_ref, vnet_our = _matched_mlps([OBS_DIM, 8, 8, 1], seed=12)
    torch.manual_seed(13)
    log_std_ref = nn.Parameter(-0.5 * torch.ones(ACT_DIM))
    log_std_our = nn.Parameter(-0.5 * torch.ones(ACT_DIM))

    if algo_name == "TRPO":
        # reference run_trpo.py:31-36 — CG optimizer over network params
        p_opt_ref = ref["ConjugateGradientOptimizer"](params=pnet_ref.parameters())
        from rl_replicas_amd.optimizers import ConjugateGradientOptimizer as OurCGO

        p_opt_our = OurCGO(params=pnet_our.parameters())
    else:
        # include log_std so its gradient path is exercised
        p_opt_ref = torch.optim.Adam(
            list(pnet_ref.parameters()) + [log_std_ref], lr=3e-4
        )
        p_opt_our = torch.optim.Adam(
            list(pnet_our.parameters()) + [log_std_our], lr=3e-4
        )

    policy_ref = ref["GaussianPolicy"](pnet_ref, p_opt_ref, log_std_ref)
    policy_our = OurGaussianPolicy(pnet_our, p_opt_our, log_std_our)
    vf_ref = ref["ValueFunction"](
        vnet_ref, torch.optim.Adam(vnet_ref.parameters(), lr=1e-3)
    )
    vf_our = OurValueFunction(
        vnet_our, torch.optim.Adam(vnet_our.parameters(), lr=1e-3)
    )

    kwargs = {}
    if algo_name == "PPO":
        kwargs = dict(num_policy_gradients=5, num_value_gradients=5)
    else:
        kwargs = dict(num_value_gradients=5)
    ref_cls = ref[algo_name]
    our_cls = {"VPG": OurVPG, "PPO": OurPPO, "TRPO": OurTRPO}[algo_name]
    algo_ref = ref_cls(policy_ref, vf_ref, fake_env, None, **kwargs)
    algo_our = our_cls(policy_our, vf_our, fake_env, None, **kwargs)
    for a in (algo_ref, algo_our):
        a.metrics_manager = FakeMetricsManager()
        a.current_total_steps = 0
        a.current_total_episodes = 0
    return algo_ref, algo_our


@pytest.mark.parametrize("algo_name", ["VPG", "PPO", "TRPO"])
def test_onpolicy_train_equivalence(algo_name):
    # TRPO's CG solves a damped Fisher system: with fewer samples than
    # policy params the Fisher is singular and CG amplifies fp noise by
    # ~1/damping, so the TRPO case uses a batch large enough for a
    # well-conditioned Fisher (sample count x act_dim >> 114 params)
    lengths = [40, 50, 30, 45] if algo_name == "TRPO" else None
    rollout = _make_rollout(np.random.default_rng(42), lengths=lengths)
    exp_ref, exp_our = _experience_pair(rollout)
    algo_ref, algo_our = _build_onpolicy_pair(algo_name)
    init = {
        k: v.clone() for k, v in algo_ref.policy.network.state_dict().items()
    }

    algo_ref.train(exp_ref)
    algo_our.train(exp_our)

    if algo_name == "TRPO":
        # Both stacks must take the SAME accepted trust-region step
        # (direction + length).  Element tolerances don't fit here: 10 CG
        # iterations on the damped (1e-5) Fisher amplify fp noise to
        # ~0.3% of the step vector, so compare the steps as vectors.
        step_ref = torch.cat(
            [
                (v - init[k]).flatten()
                for k, v in algo_ref.policy.network.state_dict().items()
            ]
        )
        step_our = torch.cat(
            [
                (v - init[k]).flatten()
                for k, v in algo_our.policy.network.state_dict().items()
            ]
        )
        cos = torch.dot(step_ref, step_our) / (step_ref.norm() * step_our.norm())
        assert float(cos) > 0.999, f"CG step direction diverged: cos={float(cos)}"
        ratio = float(step_our.norm() / step_ref.norm())
        assert abs(ratio - 1.0) < 0.02, f"accepted step length diverged: {ratio}"
    else:
        _assert_modules_close(
            algo_ref.policy.network, algo_our.policy.network,
            label=f"{algo_name}.policy"
        )
        torch.testing.assert_close(
            algo_our.policy.log_std, algo_ref.policy.log_std, rtol=1e-4, atol=1e-6
        )
    _assert_modules_close(
        algo_ref.value_function.network,
        algo_our.value_function.network,
        label=f"{algo_name}.value",
    )
    if hasattr(algo_ref, "old_policy") and algo_name != "TRPO":
        # (TRPO's old_policy is synced to the post-step policy, so it
        # carries the same CG fp noise compared above as a step vector)
        _assert_modules_close(
            algo_ref.old_policy.network,
            algo_our.old_policy.network,
            label=f"{algo_name}.old_policy",
        )

    ref_m = algo_ref.metrics_manager.scalars
    our_m = algo_our.metrics_manager.scalars
    for tag in ("policy/loss", "value_function/average_loss", "policy/avarage_entropy"):
        if tag in ref_m and tag in our_m:
            assert our_m[tag] == pytest.approx(ref_m[tag], rel=1e-3, abs=1e-6), tag


def test_ppo_kl_and_early_stop_metric_equivalence():
    """PPO's approx-KL drives early stop (reference ppo.py:173-181): the
    recorded kl_divergence after the epoch must match."""
    rollout = _make_rollout(np.random.default_rng(7))
    exp_ref, exp_our = _experience_pair(rollout)
    algo_ref, algo_our = _build_onpolicy_pair("PPO")
    algo_ref.train(exp_ref)
    algo_our.train(exp_our)
    ref_kl = algo_ref.metrics_manager.scalars["policy/kl_divergence"]
    our_kl = algo_our.metrics_manager.scalars["policy/kl_divergence"]
    assert our_kl == pytest.approx(ref_kl, rel=1e-3, abs=1e-7)


# ---------------------------------------------------------------------------
# CG optimizer: identical closures -> identical accepted step
# ---------------------------------------------------------------------------
def test_cg_optimizer_step_equivalence():
    ref = load_reference()
    from rl_replicas_amd.optimizers import ConjugateGradientOptimizer as OurCGO

    def build(cgo_cls):
        torch.manual_seed(21)
        net = nn.Sequential(nn.Linear(4, 6), nn.Tanh(), nn.Linear(6, 2))
        opt = cgo_cls(params=net.parameters())
        torch.manual_seed(22)
        obs = torch.randn(32, 4)
        target = torch.randn(32, 2)
        anchor = [p.detach().clone() for p in net.parameters()]

        def loss_fn():
            return ((net(obs) - target) ** 2).mean()

        def kl_fn():
            # zero value/grad at the current params, positive curvature:
            # a valid trust-region constraint surrogate
            return sum(((p - a) ** 2).sum() for p, a in zip(net.parameters(), anchor))

        loss = loss_fn()
        opt.zero_grad()
        loss.backward()
        opt.step(loss_fn, kl_fn)
        return net

    net_ref = build(ref["ConjugateGradientOptimizer"])
    net_our = build(OurCGO)
    for (kr, pr), (ko, po) in zip(
        net_ref.state_dict().items(), net_our.state_dict().items()
    ):
        assert kr == ko
        torch.testing.assert_close(po, pr, rtol=1e-5, atol=1e-7, msg=kr)


# ---------------------------------------------------------------------------
# off-policy: full train(buffer, n, mb) equivalence under matched RNG
# ---------------------------------------------------------------------------
def _fill_buffers(rollout, obs_dim, act_dim):
    ref = load_reference()
    from rl_replicas_amd.replay_buffer import ReplayBuffer as OurReplayBuffer

    exp_ref, exp_our = _experience_pair(rollout)
    buf_ref = ref["ReplayBuffer"](1000)
    buf_our = OurReplayBuffer(1000)
    buf_ref.add_experience(exp_ref)
    buf_our.add_experience(exp_our)
    return buf_ref, buf_our


def _build_offpolicy_pair(algo_name: str, buf_ref=None, buf_our=None):
    ref = load_reference()
    from rl_replicas_amd import envs as our_envs
    from rl_replicas_amd.algorithms import DDPG as OurDDPG
    from rl_replicas_amd.algorithms import TD3 as OurTD3
    from rl_replicas_amd.evaluator import Evaluator as OurEvaluator
    from rl_replicas_amd.policies import DeterministicPolicy as OurDetPolicy
    from rl_replicas_amd.q_function import QFunction as OurQFunction

    # Pendulum shapes: our off-policy ctor builds its evaluation env via
    # the native registry (off_policy.py), the reference via the stubbed
    # gym.make
    obs_dim, act_dim, limit = 3, 1, 2.0
    env_our = our_envs.make("Pendulum-v1")
    env_ref = make_fake_gym_env("Pendulum-v1", obs_dim, act_dim, limit)

    pnet_ref, pnet_our = _matched_mlps(
        [obs_dim, 16, 16, act_dim],
        seed=31,
        activation_function=nn.ReLU,
        output_activation_function=nn.Tanh,
    )
    policy_ref = ref["DeterministicPolicy"](
        pnet_ref, torch.optim.Adam(pnet_ref.parameters(), lr=1e-3)
    )
    policy_our = OurDetPolicy(
        pnet_our, torch.optim.Adam(pnet_our.parameters(), lr=1e-3)
    )

    def q_pair(seed):
        qnet_ref, qnet_our = _matched_mlps(
            [obs_dim + act_dim, 16, 16, 1], seed=seed, activation_function=nn.ReLU
        )
        return (
            ref["QFunction"](qnet_ref, torch.optim.Adam(qnet_ref.parameters(), lr=1e-3)),
            OurQFunction(qnet_our, torch.optim.Adam(qnet_our.parameters(), lr=1e-3)),
        )

    q1_ref, q1_our = q_pair(32)
    common_ref = dict(env=env_ref, sampler=None, evaluator=None)
    common_our = dict(env=env_our, sampler=None, evaluator=OurEvaluator())

    if algo_name == "DDPG":
        algo_ref = ref["DDPG"](
            policy_ref, None, q1_ref, replay_buffer=buf_ref, **common_ref
        )
        algo_our = OurDDPG(
            policy_our, None, q1_our, replay_buffer=buf_our, **common_our
        )
    else:
        q2_ref, q2_our = q_pair(33)
        algo_ref = ref["TD3"](
            policy_ref, None, q1_ref, q2_ref, replay_buffer=buf_ref, **common_ref
        )
        algo_our = OurTD3(
            policy_our, None, q1_our, q2_our, replay_buffer=buf_our, **common_our
        )
    for a in (algo_ref, algo_our):
        a.metrics_manager = FakeMetricsManager()
        a.current_total_steps = 0
    return algo_ref, algo_our


@pytest.mark.parametrize("algo_name", ["DDPG", "TD3"])
def test_offpolicy_train_equivalence(algo_name):
    rollout = _make_rollout(np.random.default_rng(9), obs_dim=3, act_dim=1)
    buf_ref, buf_our = _fill_buffers(rollout, 3, 1)
    algo_ref, algo_our = _build_offpolicy_pair(algo_name, buf_ref, buf_our)

    # matched RNG streams: minibatch indices come from np.random, TD3
    # smoothing noise from torch RNG, one draw per train iteration in
    # both stacks
    np.random.seed(123)
    torch.manual_seed(456)
    algo_ref.train(buf_ref, num_train_steps=6, minibatch_size=32)
    np.random.seed(123)
    torch.manual_seed(456)
    algo_our.train(buf_our, num_train_steps=6, minibatch_size=32)

    _assert_modules_close(
        algo_ref.policy.network, algo_our.policy.network, label=f"{algo_name}.policy"
    )
    _assert_modules_close(
        algo_ref.target_policy.network,
        algo_our.target_policy.network,
        label=f"{algo_name}.target_policy",
    )
    if algo_name == "DDPG":
        pairs = [("q_function", "q_function"), ("target_q_function", "target_q_function")]
    else:
        pairs = [
            ("q_function_1", "q_function_1"),
            ("q_function_2", "q_function_2"),
            ("target_q_function_1", "target_q_function_1"),
            ("target_q_function_2", "target_q_function_2"),
        ]
    for ref_name, our_name in pairs:
        _assert_modules_close(
            getattr(algo_ref, ref_name).network,
            getattr(algo_our, our_name).network,
            label=f"{algo_name}.{ref_name}",
        )

    ref_m = algo_ref.metrics_manager.scalars
    our_m = algo_our.metrics_manager.scalars
    shared = set(ref_m) & set(our_m)
    assert shared, "no common metric tags recorded"
    for tag in shared:
        assert our_m[tag] == pytest.approx(ref_m[tag], rel=1e-3, abs=1e-5), tag


def test_ppo_categorical_train_equivalence():
    """PPO with a CategoricalPolicy (the reference's discrete-action
    config, reference categorical_policy.py:8-32) — full train()
    equivalence on identical integer-action rollouts."""
    ref = load_reference()
    from rl_replicas_amd.algorithms import PPO as OurPPO
    from rl_replicas_amd.policies import CategoricalPolicy as OurCategorical
    from rl_replicas_amd.value_function import ValueFunction as OurValueFunction

    n_act = 3
    fake_env = make_fake_gym_env("FakeDiscrete-v0", OBS_DIM, n_act, 1.0)
    pnet_ref, pnet_our = _matched_mlps([OBS_DIM, 8, n_act], seed=21)
    vnet_ref, vnet_our = _matched_mlps([OBS_DIM, 8, 1], seed=22)

    p_ref = ref["CategoricalPolicy"](
        pnet_ref, torch.optim.Adam(pnet_ref.parameters(), lr=3e-4)
    )
    p_our = OurCategorical(
        pnet_our, torch.optim.Adam(pnet_our.parameters(), lr=3e-4)
    )
    vf_ref = ref["ValueFunction"](
        vnet_ref, torch.optim.Adam(vnet_ref.parameters(), lr=1e-3)
    )
    vf_our = OurValueFunction(
        vnet_our, torch.optim.Adam(vnet_our.parameters(), lr=1e-3)
    )

    algo_ref = ref["PPO"](p_ref, vf_ref, fake_env, None,
                          num_policy_gradients=4, num_value_gradients=4)
    algo_our = OurPPO(p_our, vf_our, fake_env, None,
                      num_policy_gradients=4, num_value_gradients=4)
    for a in (algo_ref, algo_our):
        a.metrics_manager = FakeMetricsManager()
        a.current_total_steps = 0
        a.current_total_episodes = 0

    rng = np.random.default_rng(31)
    rollout = _make_rollout(rng, act_dim=1)
    for ep in rollout["actions"]:  # integer actions in [0, n_act)
        for i in range(len(ep)):
            ep[i] = np.array(float(rng.integers(0, n_act)), dtype=np.float32)
    exp_ref, exp_our = _experience_pair(rollout)
    algo_ref.train(exp_ref)
    algo_our.train(exp_our)

    _assert_modules_close(algo_ref.policy.network, algo_our.policy.network,
                          label="PPO-cat.policy")
    _assert_modules_close(algo_ref.value_function.network,
                          algo_our.value_function.network, label="PPO-cat.value")
    ref_m = algo_ref.metrics_manager.scalars
    our_m = algo_our.metrics_manager.scalars
    assert our_m["policy/kl_divergence"] == pytest.approx(
        ref_m["policy/kl_divergence"], rel=1e-3, abs=1e-7
    )
