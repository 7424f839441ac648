"""CPU fallbacks of the round-2 ops + small helpers.

The CPU implementations are the numerics oracles for the HIP kernels
(ops/__init__.py contract); these pin their math directly.
"""
import numpy as np
import pytest
import torch

from rl_replicas_amd import ops


def test_q_target_min2_cpu():
    r = torch.randn(100)
    d = (torch.rand(100) < 0.2).float()
    q1 = torch.randn(100)
    q2 = torch.randn(100)
    out = ops.q_target_min2(r, d, q1, q2, 0.99)
    torch.testing.assert_close(out, r + 0.99 * (1 - d) * torch.min(q1, q2))


def test_td3_smooth_cpu_matches_reference_chain():
    """The CPU path must consume torch RNG exactly like the reference's
    randn_like chain (td3.py:325-341) — the equivalence suite depends
    on it."""
    a = torch.randn(64, 4)
    torch.manual_seed(7)
    got = ops.td3_smooth(a, 0.2, 0.5, 1.0, 0, 0)
    torch.manual_seed(7)
    eps = torch.clamp(0.2 * torch.randn_like(a), -0.5, 0.5)
    want = torch.clamp(a + eps, -1.0, 1.0)
    torch.testing.assert_close(got, want)
    assert got.min() >= -1.0 and got.max() <= 1.0


def test_adam_arg_lists_layout_and_group():
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.ops.fused_adam import adam_arg_lists

    net = MLP([4, 8, 2])
    log_std = torch.nn.Parameter(torch.zeros(2))
    opt = torch.optim.Adam(list(net.parameters()) + [log_std], lr=3e-4)
    weights = [m.weight for m in net.network if isinstance(m, torch.nn.Linear)]
    biases = [m.bias for m in net.network if isinstance(m, torch.nn.Linear)]
    m, v, step0, hp = adam_arg_lists(opt, weights, biases)
    assert len(m) == len(v) == 2 * len(weights)
    # ordering contract: weights then biases
    for i, w in enumerate(weights):
        assert m[i].shape == w.shape
    for i, b in enumerate(biases):
        assert m[len(weights) + i].shape == b.shape
    assert float(step0) == 0.0
    assert hp[0] == pytest.approx(3e-4)  # lr from the owning group


def test_gather_global_batch_single_process_identity():
    from rl_replicas_amd.parallel.ddp import gather_global_batch

    obs = torch.randn(10, 3)
    act = torch.randn(10, 2)
    adv = torch.randn(10)
    ret = torch.randn(10)
    o, a, d, r = gather_global_batch(obs, act, adv, ret)
    assert torch.equal(o, obs) and torch.equal(a, act)
    assert torch.equal(d, adv) and torch.equal(r, ret)


def test_replay_gather_cpu_unavailable_contract():
    """gather_minibatch_fused is GPU-only by design; the CPU path keeps
    the reference numpy-RNG sampling (sample_minibatch_tensors)."""
    from rl_replicas_amd.replay_buffer import ReplayBuffer

    buf = ReplayBuffer(64)
    from rl_replicas_amd.experience import Experience

    rng = np.random.default_rng(0)
    exp = Experience(
        [[rng.normal(size=3).astype(np.float32) for _ in range(5)]],
        [[rng.normal(size=1).astype(np.float32) for _ in range(5)]],
        [[1.0] * 5],
        [rng.normal(size=3).astype(np.float32)],
        [[False] * 4 + [True]],
        [5.0],
        [5],
    )
    buf.add_experience(exp)
    np.random.seed(3)
    mb = buf.sample_minibatch_tensors(8)
    np.random.seed(3)
    idx = np.random.randint(0, 5, 8)
    torch.testing.assert_close(
        mb["rewards"], buf._storage["rewards"][torch.as_tensor(idx)]
    )
    with pytest.raises(AssertionError):
        buf.gather_minibatch_fused(8, seed=0, offset=0)  # CPU buffer
