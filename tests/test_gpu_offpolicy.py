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


# ---------------------------------------------------------------------------
# off-policy kernels (offpolicy_kernels.hip)
# ---------------------------------------------------------------------------
def test_q_target_min2_matches_eager(ext):
    n = 1000
    r = torch.randn(n, device="cuda")
    d = (torch.rand(n, device="cuda") < 0.1).float()
    q1 = torch.randn(n, device="cuda")
    q2 = torch.randn(n, device="cuda")
    out = ext.q_target_min2(r, d, q1, q2, 0.99)
    oracle = r + 0.99 * (1 - d) * torch.min(q1, q2)
    torch.testing.assert_close(out, oracle, rtol=1e-6, atol=1e-7)


def test_td3_smooth_bounds_and_determinism(ext):
    a = 3.0 * torch.randn(512, 6, device="cuda")
    scale, clip, limit = 0.2, 0.5, 1.0
    out1 = ext.td3_smooth(a, 42, 7, scale, clip, limit)
    out2 = ext.td3_smooth(a, 42, 7, scale, clip, limit)
    torch.testing.assert_close(out1, out2, rtol=0.0, atol=0.0)  # stateless
    out3 = ext.td3_smooth(a, 42, 8, scale, clip, limit)
    assert not torch.equal(out1, out3)  # offset advances the stream
    assert out1.min() >= -limit and out1.max() <= limit
    # scale=0 reduces to the pure clamp
    out0 = ext.td3_smooth(a, 42, 7, 0.0, clip, limit)
    torch.testing.assert_close(out0, torch.clamp(a, -limit, limit), rtol=0.0, atol=0.0)
    # noise statistics: clipped normal, std ~ scale for clip >> scale
    wide = ext.td3_smooth(torch.zeros(200_000, 1, device="cuda"), 3, 1, 0.2, 10.0, 100.0)
    assert abs(float(wide.mean())) < 0.005
    assert abs(float(wide.std()) - 0.2) < 0.005


def test_replay_gather_matches_storage(ext):
    from rl_replicas_amd.replay_buffer import ReplayBuffer

    O, A, n = 5, 3, 400
    buf = ReplayBuffer(1024, device="cuda")
    buf._allocate(O, (A,))
    st = buf._storage
    # index-decodable observations: obs[i, 0] = i
    st["observations"][:n] = torch.arange(n, device="cuda").float()[:, None].repeat(1, O)
    st["observations"][:n, 1:] = torch.randn(n, O - 1, device="cuda")
    st["observations"][:n, 0] = torch.arange(n, device="cuda").float()
    st["actions"][:n] = torch.randn(n, A, device="cuda")
    st["rewards"][:n] = torch.randn(n, device="cuda")
    st["next_observations"][:n] = torch.randn(n, O, device="cuda")
    st["dones"][:n] = (torch.rand(n, device="cuda") < 0.5).float()
    buf.current_size = n
    buf._sync_size_dev()

    mb = buf.gather_minibatch_fused(64, seed=9, offset=0)
    idx = mb["observations"][:, 0].long()
    assert idx.min() >= 0 and idx.max() < n
    assert len(torch.unique(idx)) > 1  # actually random
    torch.testing.assert_close(mb["observations"], st["observations"][idx], rtol=0, atol=0)
    torch.testing.assert_close(mb["qin"][:, :O], st["observations"][idx], rtol=0, atol=0)
    torch.testing.assert_close(mb["qin"][:, O:], st["actions"][idx], rtol=0, atol=0)
    torch.testing.assert_close(mb["rewards"], st["rewards"][idx], rtol=0, atol=0)
    torch.testing.assert_close(mb["next_observations"], st["next_observations"][idx], rtol=0, atol=0)
    torch.testing.assert_close(mb["dones"], st["dones"][idx], rtol=0, atol=0)
    # stateless determinism + stream separation
    mb2 = buf.gather_minibatch_fused(64, seed=9, offset=0)
    torch.testing.assert_close(mb["qin"], mb2["qin"], rtol=0, atol=0)
    mb3 = buf.gather_minibatch_fused(64, seed=9, offset=1)
    assert not torch.equal(mb["qin"], mb3["qin"])
