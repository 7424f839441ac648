"""Fused off-policy steps vs torch-autograd oracles (gpu-marked)."""
import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from rl_replicas_amd import ops

    assert ops.hip_available()
    return ops._load_extension()


def make_pair(sizes, acts, seed=0):
    from rl_replicas_amd.networks import MLP

    torch.manual_seed(seed)
    a = (MLP(sizes) if acts is None else MLP(sizes, *acts)).to("cuda")
    b = (MLP(sizes) if acts is None else MLP(sizes, *acts)).to("cuda")
    b.load_state_dict(a.state_dict())
    return a, b


def test_q_step_matches_eager(ext):
    from rl_replicas_amd.ops import fused_offpolicy as fop
    from rl_replicas_amd.q_function import QFunction

    qn_f, qn_e = make_pair([23, 256, 256, 1], (nn.ReLU, nn.Identity))
    qf = QFunction(qn_f, torch.optim.Adam(qn_f.parameters(), lr=1e-3))
    qe = QFunction(qn_e, torch.optim.Adam(qn_e.parameters(), lr=1e-3))

    obs = torch.randn(100, 17, device="cuda")
    act = torch.randn(100, 6, device="cuda")
    targets = torch.randn(100, device="cuda")

    loss_f = fop.q_step(qf, obs, act, targets, lambda m: None)

    q = qe(obs, act)
    loss_e = torch.nn.functional.mse_loss(q, targets)
    qe.optimizer.zero_grad()
    loss_e.backward()
    qe.optimizer.step()

    torch.testing.assert_close(loss_f, loss_e.detach(), rtol=1e-5, atol=1e-6)
    for p_f, p_e in zip(qf.parameters(), qe.parameters()):
        torch.testing.assert_close(p_f, p_e, rtol=1e-4, atol=1e-6)


def test_policy_step_matches_eager(ext):
    from rl_replicas_amd.ops import fused_offpolicy as fop
    from rl_replicas_amd.policies import DeterministicPolicy
    from rl_replicas_amd.q_function import QFunction

    pn_f, pn_e = make_pair([17, 256, 256, 6], (nn.ReLU, nn.Tanh), seed=1)
    qn_f, qn_e = make_pair([23, 256, 256, 1], (nn.ReLU, nn.Identity), seed=2)
    pf = DeterministicPolicy(pn_f, torch.optim.Adam(pn_f.parameters(), lr=1e-3))
    pe = DeterministicPolicy(pn_e, torch.optim.Adam(pn_e.parameters(), lr=1e-3))
    qf = QFunction(qn_f, torch.optim.Adam(qn_f.parameters(), lr=1e-3))
    qe = QFunction(qn_e, torch.optim.Adam(qn_e.parameters(), lr=1e-3))

    obs = torch.randn(100, 17, device="cuda")

    loss_f = fop.policy_step(pf, qf, obs, lambda m: None)

    # eager oracle with critic frozen (reference semantics)
    for p in qn_e.parameters():
        p.requires_grad = False
    a = pe(obs)
    loss_e = -torch.mean(qe(obs, a))
    pe.optimizer.zero_grad()
    loss_e.backward()
    pe.optimizer.step()
    for p in qn_e.parameters():
        p.requires_grad = True

    torch.testing.assert_close(loss_f, loss_e.detach(), rtol=1e-4, atol=1e-6)
    for p_f, p_e in zip(pf.parameters(), pe.parameters()):
        torch.testing.assert_close(p_f, p_e, rtol=1e-4, atol=1e-6)
    # critic untouched by the actor step in both paths
    for p_f, p_e in zip(qf.parameters(), qe.parameters()):
        torch.testing.assert_close(p_f, p_e, rtol=0.0, atol=0.0)
