"""Analytic FVP vs double-backward FVP (gpu-marked).

TRPO evaluates the KL Hessian at policy == old_policy, where the
Gauss-Newton/Fisher form is exact; these tests pin the analytic
implementation (ops/fused_trpo.py) against the reference-style
autograd double-backward on the same closures."""
import copy

import pytest
import torch
import torch.nn as nn
from torch.distributions import kl_divergence

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def require_hip():
    from rl_replicas_amd import ops

    assert ops.hip_available()


def autograd_fvp(policy, old_policy, obs, params, damping):
    from rl_replicas_amd.optimizers.conjugate_gradient_optimizer import (
        ConjugateGradientOptimizer,
        _flatten,
    )

    def kl_fn():
        dist = policy(obs)
        with torch.no_grad():
            old = old_policy(obs)
        return torch.mean(kl_divergence(old, dist))

    opt = ConjugateGradientOptimizer(params, hvp_damping_coefficient=damping)
    return opt._make_fisher_vector_product(kl_fn, params)


def test_gaussian_fvp_matches_autograd():
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.ops import fused_trpo
    from rl_replicas_amd.optimizers import ConjugateGradientOptimizer
    from rl_replicas_amd.policies import GaussianPolicy

    torch.manual_seed(0)
    net = MLP([17, 64, 32, 6]).to("cuda")
    log_std = nn.Parameter(-0.4 * torch.ones(6, device="cuda"))
    params = list(net.parameters()) + [log_std]
    policy = GaussianPolicy(net, ConjugateGradientOptimizer(params), log_std)
    net.fused_training = False  # autograd closures need eager forwards
    old_policy = copy.deepcopy(policy)
    obs = torch.randn(2000, 17, device="cuda")

    # grads must exist for the optimizer param filter
    loss = policy(obs).log_prob(torch.randn(2000, 6, device="cuda")).mean()
    loss.backward()

    damping = 1e-5
    fvp_ref = autograd_fvp(policy, old_policy, obs, params, damping)
    fvp_ana = fused_trpo.make_fvp(policy, obs, damping)
    assert fvp_ana is not None

    torch.manual_seed(1)
    total = sum(p.numel() for p in params)
    for _ in range(3):
        v = torch.randn(total, device="cuda")
        hv_ref = fvp_ref(v)
        hv_ana = fvp_ana(v)
        torch.testing.assert_close(hv_ana, hv_ref, rtol=2e-3, atol=1e-5)


def test_categorical_fvp_matches_autograd():
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.ops import fused_trpo
    from rl_replicas_amd.optimizers import ConjugateGradientOptimizer
    from rl_replicas_amd.policies import CategoricalPolicy

    torch.manual_seed(0)
    net = MLP([4, 64, 32, 2]).to("cuda")
    params = list(net.parameters())
    policy = CategoricalPolicy(net, ConjugateGradientOptimizer(params))
    net.fused_training = False
    old_policy = copy.deepcopy(policy)
    obs = torch.randn(1000, 4, device="cuda")

    acts = torch.randint(0, 2, (1000,), device="cuda")
    loss = policy(obs).log_prob(acts).mean()
    loss.backward()

    damping = 1e-5
    fvp_ref = autograd_fvp(policy, old_policy, obs, params, damping)
    fvp_ana = fused_trpo.make_fvp(policy, obs, damping)
    assert fvp_ana is not None

    total = sum(p.numel() for p in params)
    for _ in range(3):
        v = torch.randn(total, device="cuda")
        torch.testing.assert_close(fvp_ana(v), fvp_ref(v), rtol=2e-3, atol=1e-5)


def test_trpo_gpu_uses_analytic_fvp_and_learns(tmp_path):
    import numpy as np

    from rl_replicas_amd import envs, ops
    from rl_replicas_amd.algorithms import TRPO
    from rl_replicas_amd.evaluator import Evaluator
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.optimizers import ConjugateGradientOptimizer
    from rl_replicas_amd.policies import CategoricalPolicy
    from rl_replicas_amd.samplers import VectorSampler
    from rl_replicas_amd.value_function import ValueFunction

    torch.manual_seed(0)
    venv = envs.VectorEnv("CartPole-v1", num_envs=10)
    pnet = MLP([4, 64, 32, 2]).to("cuda")
    policy = CategoricalPolicy(pnet, ConjugateGradientOptimizer(pnet.parameters()))
    vnet = MLP([4, 64, 32, 1]).to("cuda")
    vf = ValueFunction(vnet, ops.make_adam(vnet.parameters(), lr=1e-3))
    model = TRPO(policy, vf, venv, VectorSampler(venv, seed=0))
    model.learn(num_epochs=5, batch_size=500, output_dir=str(tmp_path))
    returns, _ = Evaluator(seed=0).evaluate(policy, envs.make("CartPole-v1"), 3)
    assert np.mean(returns) > 30.0


def test_captured_cg_matches_eager_solve(monkeypatch):
    """The hipGraph-captured CG solve (CapturableFVP buffers + masked
    fixed-iteration loop, ONE host sync per solve) must produce the
    same trust-region step as the eager device loop."""
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.ops import fused_trpo
    from rl_replicas_amd.optimizers import ConjugateGradientOptimizer
    from rl_replicas_amd.policies import GaussianPolicy

    def one_solve(graphs: bool):
        if graphs:
            monkeypatch.delenv("RL_REPLICAS_AMD_DISABLE_GRAPHS", raising=False)
        else:
            monkeypatch.setenv("RL_REPLICAS_AMD_DISABLE_GRAPHS", "1")
        torch.manual_seed(3)
        net = MLP([17, 64, 32, 6]).to("cuda")
        log_std = nn.Parameter(-0.5 * torch.ones(6, device="cuda"))
        params = list(net.parameters()) + [log_std]
        opt = ConjugateGradientOptimizer(params)
        policy = GaussianPolicy(net, opt, log_std)
        old_policy = copy.deepcopy(policy)
        torch.manual_seed(4)
        obs = torch.randn(1000, 17, device="cuda")
        actions = torch.randn(1000, 6, device="cuda")
        adv = torch.randn(1000, device="cuda")

        def loss_fn():
            logp = policy(obs).log_prob(actions)
            with torch.no_grad():
                old_logp = old_policy(obs).log_prob(actions)
            return -torch.mean(torch.exp(logp - old_logp) * adv)

        def kl_fn():
            dist = policy(obs)
            with torch.no_grad():
                old = old_policy(obs)
            return torch.mean(kl_divergence(old, dist))

        net.fused_training = False
        loss = loss_fn()
        opt.zero_grad()
        loss.backward()
        fvp = fused_trpo.make_fvp(policy, obs, opt.hvp_damping_coefficient)
        assert fvp is not None
        if graphs:
            assert getattr(fvp, "graph_key", None) is not None
        opt.step(loss_fn, kl_fn, fisher_vector_product=fvp)
        torch.cuda.synchronize()
        return [p.detach().clone() for p in params]

    eager = one_solve(graphs=False)
    graphed = one_solve(graphs=True)
    for p_e, p_g in zip(eager, graphed):
        torch.testing.assert_close(p_g, p_e, rtol=1e-5, atol=1e-7)


def test_captured_cg_reused_across_epochs(monkeypatch):
    """Second epoch reuses the captured solve (same FVP object, refreshed
    buffers) and still matches the eager result."""
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.ops import fused_trpo
    from rl_replicas_amd.optimizers import ConjugateGradientOptimizer
    from rl_replicas_amd.policies import GaussianPolicy

    monkeypatch.delenv("RL_REPLICAS_AMD_DISABLE_GRAPHS", raising=False)
    torch.manual_seed(9)
    net = MLP([17, 64, 32, 6]).to("cuda")
    log_std = nn.Parameter(-0.5 * torch.ones(6, device="cuda"))
    params = list(net.parameters()) + [log_std]
    opt = ConjugateGradientOptimizer(params)
    policy = GaussianPolicy(net, opt, log_std)
    net.fused_training = False

    for epoch in range(2):
        torch.manual_seed(100 + epoch)
        obs = torch.randn(500, 17, device="cuda")
        v = torch.randn(sum(p.numel() for p in params), device="cuda")
        loss = policy(obs).log_prob(torch.randn(500, 6, device="cuda")).mean()
        opt.zero_grad()
        loss.backward()
        fvp = fused_trpo.make_fvp(policy, obs, 1e-5)
        if epoch == 0:
            first_fvp = fvp
        else:
            assert fvp is first_fvp  # policy-cached, buffers refreshed
        # captured solve vs eager masked loop on the same refreshed FVP
        b = torch.cat([p.grad.reshape(-1) for p in params]).detach()
        solver = opt._get_captured_cg(fvp, b)
        assert solver is not None
        direction_g, quad_g = solver.run(b)
        direction_e = torch.nan_to_num(opt._conjugate_gradient(fvp, b), nan=0.0)
        quad_e = torch.dot(direction_e, fvp(direction_e)) + 1e-8
        torch.testing.assert_close(direction_g, direction_e, rtol=1e-5, atol=1e-7)
        torch.testing.assert_close(quad_g, quad_e, rtol=1e-5, atol=1e-8)
