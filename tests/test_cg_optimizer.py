"""ConjugateGradientOptimizer unit tests (the reference ships none —
SURVEY.md §4 calls these out as a gap to close)."""
import numpy as np
import torch
import torch.nn as nn

from rl_replicas_amd.optimizers import ConjugateGradientOptimizer


def test_cg_solves_spd_system():
    """The internal CG solves Hx=b against numpy.linalg.solve."""
    n = 12
    rng = np.random.default_rng(0)
    A = rng.standard_normal((n, n))
    H = A @ A.T + n * np.eye(n)  # SPD, well-conditioned
    b = rng.standard_normal(n)

    H_t = torch.as_tensor(H, dtype=torch.float64)
    b_t = torch.as_tensor(b, dtype=torch.float64)
    dummy = nn.Parameter(torch.zeros(n, dtype=torch.float64))
    opt = ConjugateGradientOptimizer([dummy], n_conjugate_gradients=50)
    x = opt._conjugate_gradient(lambda v: H_t @ v, b_t)
    np.testing.assert_allclose(x.numpy(), np.linalg.solve(H, b), rtol=1e-4, atol=1e-7)


def _quadratic_problem():
    """min_w loss(w) = 0.5 w^T Q w - c^T w with 'KL' = 0.5 (w-w0)^T P (w-w0)."""
    torch.manual_seed(0)
    n = 6
    Qm = torch.randn(n, n)
    Q = Qm @ Qm.T + n * torch.eye(n)
    c = torch.randn(n)
    Pm = torch.randn(n, n)
    P = Pm @ Pm.T + n * torch.eye(n)
    w = nn.Parameter(torch.zeros(n))
    w0 = w.detach().clone()

    def loss_fn():
        return 0.5 * w @ Q @ w - c @ w

    def kl_fn():
        d = w - w0
        return 0.5 * d @ P @ d

    return w, w0, Q, c, P, loss_fn, kl_fn


def test_step_direction_and_trust_region():
    """After one step: the update is along P^-1 g (damping ~0) and the
    quadratic constraint is within max_constraint."""
    w, w0, Q, c, P, loss_fn, kl_fn = _quadratic_problem()
    max_kl = 0.01
    opt = ConjugateGradientOptimizer([w], max_constraint=max_kl, n_conjugate_gradients=50)

    loss = loss_fn()
    loss.backward()
    g = w.grad.detach().clone()
    opt.step(loss_fn, kl_fn)

    step = (w0 - w.detach()).numpy()  # params moved by -ratio*beta*direction
    direction = torch.linalg.solve(P, g).numpy()
    # collinearity with P^-1 g
    cos = np.dot(step, direction) / (np.linalg.norm(step) * np.linalg.norm(direction))
    assert cos > 0.9999
    # constraint respected
    assert float(kl_fn()) <= max_kl + 1e-6
    # loss strictly improved
    assert float(loss_fn()) < float(loss)


def test_rejects_step_when_no_improvement():
    """If every candidate step violates the constraint/loss conditions,
    parameters roll back exactly (reference CGO:230-250 semantics)."""
    torch.manual_seed(0)
    w = nn.Parameter(torch.ones(3))
    w_before = w.detach().clone()
    opt = ConjugateGradientOptimizer([w], max_backtracks=3)

    def loss_fn():
        return (w**2).sum()

    def kl_fn():
        # constraint always violated -> rejection
        return torch.tensor(1e9) + 0.0 * (w**2).sum()

    loss = loss_fn()
    loss.backward()
    # kl_fn must be differentiable wrt w for the FVP; use a w-dependent one
    def kl_fn2():
        return 1e9 * ((w - w.detach().clone() + 1e-3) ** 2).sum()

    opt.step(loss_fn, kl_fn2)
    torch.testing.assert_close(w.detach(), w_before)


def test_state_roundtrip():
    w = nn.Parameter(torch.ones(2))
    opt = ConjugateGradientOptimizer([w], max_constraint=0.05, n_conjugate_gradients=7)
    st = opt.state
    assert st["max_constraint"] == 0.05 and st["n_conjugate_gradients"] == 7
    opt2 = ConjugateGradientOptimizer([w])
    opt2.state = st
    assert opt2.max_constraint == 0.05 and opt2.n_conjugate_gradients == 7


def test_unflatten_tensor():
    w = nn.Parameter(torch.ones(2))
    opt = ConjugateGradientOptimizer([w])
    flat = torch.arange(10, dtype=torch.float32)
    parts = opt.unflatten_tensor(flat, [torch.Size([2, 3]), torch.Size([4])])
    assert parts[0].shape == (2, 3) and parts[1].shape == (4,)
    torch.testing.assert_close(parts[0].flatten(), flat[:6])
