"""Numeric-utility tests: the reference had none for utils (SURVEY.md
§4) — these pin the scan/GAE/normalize/polyak semantics against
independent oracles (scipy lfilter, explicit formulas)."""
import numpy as np
import pytest
import scipy.signal
import torch
import torch.nn as nn

from rl_replicas_amd import ops
from rl_replicas_amd.networks import MLP
from rl_replicas_amd.utils import (
    add_noise_to_get_action,
    bootstrap_rewards_with_last_values,
    compute_values,
    discounted_cumulative_sums,
    gae,
    normalize_tensor,
    polyak_average,
)
from rl_replicas_amd.value_function import ValueFunction


class TestDiscountedCumulativeSums:
    def test_matches_scipy_lfilter(self):
        x = np.random.randn(1000).astype(np.float64)
        for discount in (0.9, 0.99, 1.0):
            expected = scipy.signal.lfilter([1], [1, -discount], x[::-1])[::-1]
            np.testing.assert_allclose(discounted_cumulative_sums(x, discount), expected, rtol=1e-10)

    def test_closed_form(self):
        out = discounted_cumulative_sums(np.array([1.0, 1.0, 1.0]), 0.5)
        np.testing.assert_allclose(out, [1.75, 1.5, 1.0])


class TestGae:
    def test_matches_manual_formula(self):
        L = 50
        gamma, lam = 0.99, 0.97
        rewards = np.random.randn(L + 1)
        values = np.random.randn(L + 1)
        deltas = rewards[:-1] + gamma * values[1:] - values[:-1]
        expected = scipy.signal.lfilter([1], [1, -gamma * lam], deltas[::-1])[::-1]
        np.testing.assert_allclose(gae(rewards, gamma, values, lam), expected, rtol=1e-10)


class TestSegmentedGaeOracle:
    """ops.gae_advantages_and_returns == the reference per-episode
    pipeline (bootstrap -> scipy scans)."""

    def test_matches_reference_pipeline(self):
        gamma, lam = 0.99, 0.97
        lengths = [7, 13, 1, 29]
        dones = [True, False, True, False]
        rewards_eps = [np.random.randn(L) for L in lengths]
        values_eps = [np.random.randn(L) for L in lengths]
        last_values = [float(np.random.randn()) for _ in lengths]

        # reference pipeline (ppo.py:139-161 semantics)
        exp_returns, exp_advs = [], []
        for r, v, lv, d in zip(rewards_eps, values_eps, last_values, dones):
            rb = np.concatenate([r, [0.0 if d else lv]])
            exp_returns.append(discounted_cumulative_sums(rb, gamma)[:-1])
            v_full = np.concatenate([v, [lv]])
            exp_advs.append(gae(rb, gamma, v_full, lam))

        offsets = np.zeros(len(lengths) + 1, dtype=np.int64)
        np.cumsum(lengths, out=offsets[1:])
        adv, ret = ops.gae_advantages_and_returns(
            torch.as_tensor(np.concatenate(rewards_eps), dtype=torch.float32),
            torch.as_tensor(np.concatenate(values_eps), dtype=torch.float32),
            torch.as_tensor(np.asarray(last_values), dtype=torch.float32),
            torch.as_tensor(offsets),
            torch.as_tensor(np.asarray(dones)),
            gamma,
            lam,
        )
        np.testing.assert_allclose(ret.numpy(), np.concatenate(exp_returns), rtol=2e-5, atol=1e-5)
        np.testing.assert_allclose(adv.numpy(), np.concatenate(exp_advs), rtol=2e-5, atol=1e-5)


class TestBootstrap:
    def test_terminal_gets_zero_truncated_gets_value(self):
        out = bootstrap_rewards_with_last_values([[1.0, 2.0], [3.0]], [True, False], [9.0, 7.0])
        np.testing.assert_allclose(out[0], [1.0, 2.0, 0.0])
        np.testing.assert_allclose(out[1], [3.0, 7.0])


class TestNormalize:
    def test_zero_mean_unit_std(self):
        x = torch.randn(1000) * 3 + 5
        y = normalize_tensor(x)
        assert abs(float(y.mean())) < 1e-5
        assert abs(float(y.std()) - 1.0) < 1e-5

    def test_matches_reference_formula(self):
        x = torch.randn(257)
        expected = (x - torch.mean(x)) / torch.std(x)
        torch.testing.assert_close(normalize_tensor(x), expected)


class TestPolyak:
    def test_interpolation(self):
        net = nn.Linear(4, 3)
        target = nn.Linear(4, 3)
        before = [p.detach().clone() for p in target.parameters()]
        polyak_average(net.parameters(), target.parameters(), rho=0.9)
        for p, t0, t1 in zip(net.parameters(), before, target.parameters()):
            torch.testing.assert_close(t1.detach(), 0.9 * t0 + 0.1 * p.detach())


class TestComputeValues:
    def test_batched_equals_per_episode(self):
        net = MLP([4, 8, 1])
        vf = ValueFunction(net, torch.optim.Adam(net.parameters()))
        episodes = [[np.random.randn(4).astype(np.float32) for _ in range(L)] for L in (3, 6)]
        got = compute_values(episodes, vf)
        for ep, v in zip(episodes, got):
            with torch.no_grad():
                expected = vf(torch.as_tensor(np.stack(ep))).flatten().numpy()
            np.testing.assert_allclose(v, expected, rtol=1e-6)


class TestNoisedPolicy:
    def test_noise_is_clipped(self):
        from rl_replicas_amd import envs
        from rl_replicas_amd.policies import DeterministicPolicy

        space = envs.Box(-2.0, 2.0, shape=(1,))
        net = MLP([3, 4, 1], output_activation_function=nn.Tanh)
        base = DeterministicPolicy(net, torch.optim.Adam(net.parameters()))
        noised = add_noise_to_get_action(base, space, action_noise_scale=10.0)
        for _ in range(20):
            a = noised.get_action_numpy(np.random.randn(3).astype(np.float32))
            assert np.all(np.abs(a) <= 2.0)


class TestSegmentedGaeBruteForce:
    """Third independent derivation: the textbook double-sum definitions
    A_t = sum_l (gamma*lam)^l * delta_{t+l} and R_t = sum_k gamma^k r_{t+k}
    + gamma^{L-t} * bootstrap, evaluated directly in O(L^2)."""

    def test_matches_double_sum_definition(self):
        rng = np.random.default_rng(0)
        gamma, lam = 0.99, 0.97
        lengths = [5, 11, 2]
        dones = [False, True, False]
        r_eps = [rng.standard_normal(L).astype(np.float32) for L in lengths]
        v_eps = [rng.standard_normal(L).astype(np.float32) for L in lengths]
        lv = rng.standard_normal(len(lengths)).astype(np.float32)

        exp_adv, exp_ret = [], []
        for r, v, last, d in zip(r_eps, v_eps, lv, dones):
            L = len(r)
            v_full = np.concatenate([v, [last]])  # deltas always bootstrap
            deltas = r + gamma * v_full[1:] - v_full[:-1]
            boot = 0.0 if d else float(last)
            for t in range(L):
                exp_adv.append(sum((gamma * lam) ** l * deltas[t + l] for l in range(L - t)))
                exp_ret.append(
                    sum(gamma ** k * r[t + k] for k in range(L - t))
                    + gamma ** (L - t) * boot
                )

        offsets = np.zeros(len(lengths) + 1, dtype=np.int64)
        np.cumsum(lengths, out=offsets[1:])
        adv, ret = ops.gae_advantages_and_returns(
            torch.as_tensor(np.concatenate(r_eps)),
            torch.as_tensor(np.concatenate(v_eps)),
            torch.as_tensor(lv),
            torch.as_tensor(offsets),
            torch.as_tensor(np.asarray(dones)),
            gamma,
            lam,
        )
        np.testing.assert_allclose(adv.numpy(), exp_adv, rtol=2e-5, atol=2e-5)
        np.testing.assert_allclose(ret.numpy(), exp_ret, rtol=2e-5, atol=2e-5)
