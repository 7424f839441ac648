"""TRPO's trust-region optimizer: CG solve + backtracking line search.

Behavioral parity with the reference `ConjugateGradientOptimizer`
(reference: src/rl_replicas/optimizers/conjugate_gradient_optimizer.py
:24-265), including its exact robustness semantics:
NaN step directions zeroed, NaN step size -> 1.0, destructive in-place
parameter writes during the line search with full rollback on
rejection, and the strict (>=) rejection re-check after the loop.

MI355X notes: the Fisher-vector product is two autograd passes through
the fused-MLP HIP ops (each custom op implements double backward via
its composite autograd definition); the flat parameter/direction
vectors stay resident on device across all CG iterations (dot/axpy run
as device ops; for the tiny models here the host loop cost is the
launch latency, which HIP-graph capture of the CG body can amortize).
"""
from __future__ import annotations

import logging
from typing import Callable, Iterable, List, Optional, Tuple

import torch
from torch import Tensor
from torch.optim import Optimizer

try:
    from typing_extensions import TypedDict
except ImportError:  # pragma: no cover
    from typing import TypedDict  # type: ignore

logger = logging.getLogger(__name__)


State = TypedDict(
    "State",
    {
        "max_constraint": float,
        "n_conjugate_gradients": int,
        "max_backtracks": int,
        "backtrack_ratio": float,
        "hvp_damping_coefficient": float,
    },
)


def _flatten(tensors: Iterable[Tensor]) -> Tensor:
    return torch.cat([t.reshape(-1) for t in tensors])


class _CapturedCG:
    """The masked fixed-iteration CG solve + NaN guard + quadratic form
    captured as ONE hipGraph (inputs: the capture-stable `b` buffer and
    the FVP's activation buffers; outputs: direction, dHd+1e-8)."""

    def __init__(self, opt: "ConjugateGradientOptimizer", hvp, loss_grad: Tensor):
        from rl_replicas_amd.ops.fused_onpolicy import _CapturedLoop

        self.hvp = hvp
        self.b = loss_grad.detach().clone()
        outer = self

        def body():
            x = opt._conjugate_gradient(outer.hvp, outer.b)
            direction = torch.nan_to_num(x, nan=0.0)
            quad = torch.dot(direction, outer.hvp(direction)) + 1e-8
            return direction, quad

        # body is pure (reads buffers, writes fresh outputs): no state
        # to snapshot/restore around the warmup runs
        self.loop = _CapturedLoop(body, [])

    def run(self, loss_grad: Tensor):
        self.b.copy_(loss_grad)
        return self.loop.replay()


class ConjugateGradientOptimizer(Optimizer):
    """Constrained step: x = H^-1 g via CG, scaled to the KL trust region.

    H is the Hessian of the constraint (KL), g the loss gradient.
    """

    def __init__(
        self,
        params: Iterable[Tensor],
        max_constraint: float = 0.01,
        n_conjugate_gradients: int = 10,
        max_backtracks: int = 15,
        backtrack_ratio: float = 0.8,
        hvp_damping_coefficient: float = 1e-5,
    ):
        super().__init__(params, {})
        self.max_constraint = max_constraint
        self.n_conjugate_gradients = n_conjugate_gradients
        self.max_backtracks = max_backtracks
        self.backtrack_ratio = backtrack_ratio
        self.hvp_damping_coefficient = hvp_damping_coefficient

    # ------------------------------------------------------------------
    def step(
        self,
        loss_function: Callable,
        kl_divergence_function: Callable,
        fisher_vector_product: Optional[Callable] = None,
        reduce_hook: Optional[Callable[[Tensor], Tensor]] = None,
    ) -> None:  # type: ignore[override]
        """One trust-region update.

        Caller must have populated `.grad` of the params with the loss
        gradient (reference trpo.py:236-240 does `loss.backward()` then
        `optimizer.step(loss_fn, kl_fn)`).

        `fisher_vector_product` (extension): an analytic v -> (H + damping)v
        callable replacing the double-backward FVP — TRPO supplies the
        Gauss-Newton form, exact at its evaluation point where
        policy == old_policy (algorithms/trpo.py).

        `reduce_hook` (extension, data parallelism): applied to every
        FVP output and every line-search loss/KL evaluation so that ALL
        ranks solve the same global-batch trust-region problem and take
        bitwise-identical steps.  The caller pre-reduces `.grad`; mean of
        per-rank Hv == global-batch Hv at equal shard sizes, so with this
        hook the N-rank step equals the full-batch step
        (tests/test_parallel_cpu.py::_worker_trpo_sync).
        """
        params: List[Tensor] = [
            p for group in self.param_groups for p in group["params"] if p.grad is not None
        ]
        loss_grad = _flatten([p.grad for p in params]).detach()

        hvp = fisher_vector_product or self._make_fisher_vector_product(
            kl_divergence_function, params
        )
        if reduce_hook is not None:
            local_hvp, local_loss, local_kl = hvp, loss_function, kl_divergence_function
            hvp = lambda v: reduce_hook(local_hvp(v))  # noqa: E731
            loss_function = lambda: reduce_hook(local_loss())  # noqa: E731
            kl_divergence_function = lambda: reduce_hook(local_kl())  # noqa: E731

        solver = None
        if reduce_hook is None and loss_grad.is_cuda:
            solver = self._get_captured_cg(hvp, loss_grad)
        if solver is not None:
            # whole CG solve (+ nan guard + quadratic form) = ONE hipGraph
            # replay; the sqrt readback below is the solve's single sync
            direction, quad = solver.run(loss_grad)
        else:
            direction = self._conjugate_gradient(hvp, loss_grad)
            # NaN guard on the direction (reference :83)
            direction = torch.nan_to_num(direction, nan=0.0)
            # beta = sqrt(2*delta / (d^T H d)) (reference :86-90)
            quad = torch.dot(direction, hvp(direction)) + 1e-8
        step_size = float(torch.sqrt(2.0 * self.max_constraint / quad))
        if step_size != step_size:  # NaN guard (reference :92-94)
            step_size = 1.0

        self._backtracking_line_search(
            params, step_size * direction, loss_function, kl_divergence_function
        )

    # ------------------------------------------------------------------
    def _get_captured_cg(self, hvp, loss_grad: Tensor):
        """hipGraph-captured CG solve, cached across epochs.

        Only for an FVP with capture-stable buffers (fused_trpo.
        CapturableFVP exposes `graph_key`); the double-backward FVP
        builds autograd graphs and cannot be captured."""
        import os

        key = getattr(hvp, "graph_key", None)
        if key is None or os.environ.get("RL_REPLICAS_AMD_DISABLE_GRAPHS", "0") == "1":
            return None
        full_key = (key, int(loss_grad.numel()), self.n_conjugate_gradients)
        cached = getattr(self, "_cg_graph", None)
        # identity check matters: the captured graph bakes the FVP's
        # buffer POINTERS — a rebuilt FVP object (new buffers) needs a
        # fresh capture even if its key matches
        if cached is not None and cached[0] == full_key and cached[1].hvp is hvp:
            return cached[1]
        solver = _CapturedCG(self, hvp, loss_grad)
        self._cg_graph = (full_key, solver)
        return solver

    # ------------------------------------------------------------------
    def _make_fisher_vector_product(
        self, kl_function: Callable, params: List[Tensor]
    ) -> Callable[[Tensor], Tensor]:
        """FVP via double backward: v -> H v + damping*v.

        The KL gradient graph is built ONCE (create_graph=True); each
        call does one backward through it (reference :133-167).
        """
        kl = kl_function()
        kl_grads: Tuple[Tensor, ...] = torch.autograd.grad(kl, params, create_graph=True)
        flat_kl_grad = _flatten(kl_grads)

        def fvp(vector: Tensor) -> Tensor:
            gvp = torch.dot(flat_kl_grad, vector)
            hvp_list = torch.autograd.grad(gvp, params, retain_graph=True, allow_unused=True)
            hvp_list = [
                h if h is not None else torch.zeros_like(p) for h, p in zip(hvp_list, params)
            ]
            return _flatten(hvp_list) + self.hvp_damping_coefficient * vector

        return fvp

    # ------------------------------------------------------------------
    def _conjugate_gradient(
        self, hvp: Callable[[Tensor], Tensor], b: Tensor, residual_tol: float = 1e-10
    ) -> Tensor:
        """Solve H x = b (standard CG; reference :169-202) with ZERO
        host synchronization inside the loop.

        The reference's early exit (`if r_dot_r < tol: break`) forces a
        device->host readback per iteration; here the loop runs a FIXED
        n_conjugate_gradients iterations and the exit is a device-side
        mask: once the residual drops below tol (or goes NaN, which the
        reference also never recovers from), every later update
        multiplies by zero / keeps the frozen state, so the returned x
        is the one the reference would have returned — without a single
        `.item()`-class sync.  This also makes the whole solve
        hipGraph-capturable (fixed kernel sequence).
        """
        x = torch.zeros_like(b)
        r = b.clone()
        p = b.clone()
        r_dot_r = torch.dot(r, r)
        one = torch.ones((), device=b.device, dtype=b.dtype)
        zero = torch.zeros((), device=b.device, dtype=b.dtype)
        # iteration 0 always runs (the reference checks AFTER the update)
        active = torch.ones((), device=b.device, dtype=torch.bool)
        for _ in range(self.n_conjugate_gradients):
            hp = hvp(p)
            p_dot_hp = torch.dot(p, hp)
            alpha = torch.where(active, r_dot_r / p_dot_hp, zero)
            x = x + alpha * p
            r = r - alpha * hp
            new_r_dot_r = torch.dot(r, r)
            beta = new_r_dot_r / torch.where(r_dot_r == 0, one, r_dot_r)
            p = torch.where(active, r + beta * p, p)
            r_dot_r = torch.where(active, new_r_dot_r, r_dot_r)
            active = active & (r_dot_r >= residual_tol)
        return x

    # ------------------------------------------------------------------
    def _backtracking_line_search(
        self,
        params: List[Tensor],
        descent_step: Tensor,
        loss_function: Callable,
        constraint_function: Callable,
    ) -> None:
        """Shrink the step by `backtrack_ratio` until loss improves and
        the KL constraint holds; roll back entirely on failure
        (reference :204-250).
        """
        saved = [p.detach().clone() for p in params]
        # line-search evaluations only need VALUES (the reference builds
        # throwaway graphs here, CGO:204-250) — no_grad lets the fused
        # inference kernels run
        with torch.no_grad():
            loss_before = loss_function()

        # per-parameter views of the flat step
        numels = [p.numel() for p in params]
        step_views = [
            s.view(p.shape)
            for s, p in zip(torch.split(descent_step, numels), params)
        ]

        new_loss = loss_before
        constraint = torch.zeros((), device=descent_step.device)
        for i in range(self.max_backtracks):
            ratio = self.backtrack_ratio**i
            with torch.no_grad():
                for p, p0, s in zip(params, saved, step_views):
                    p.data.copy_(p0 - ratio * s)
                new_loss = loss_function()
                constraint = constraint_function()
            if new_loss < loss_before and constraint <= self.max_constraint:
                break

        # strict post-check: reject & roll back (reference :230-250)
        if (
            torch.isnan(new_loss)
            or torch.isnan(constraint)
            or new_loss >= loss_before
            or constraint >= self.max_constraint
        ):
            logger.warning("Line search condition violated. Rejecting the step.")
            with torch.no_grad():
                for p, p0 in zip(params, saved):
                    p.data.copy_(p0)

    # ------------------------------------------------------------------
    # state / resume (reference :100-131)
    # ------------------------------------------------------------------
    @property
    def state(self) -> State:  # type: ignore[override]
        return {
            "max_constraint": self.max_constraint,
            "n_conjugate_gradients": self.n_conjugate_gradients,
            "max_backtracks": self.max_backtracks,
            "backtrack_ratio": self.backtrack_ratio,
            "hvp_damping_coefficient": self.hvp_damping_coefficient,
        }

    @state.setter
    def state(self, state: State) -> None:
        self.max_constraint = state.get("max_constraint", 0.01)
        self.n_conjugate_gradients = state.get("n_conjugate_gradients", 10)
        self.max_backtracks = state.get("max_backtracks", 15)
        self.backtrack_ratio = state.get("backtrack_ratio", 0.8)
        self.hvp_damping_coefficient = state.get("hvp_damping_coefficient", 1e-5)

    def __setstate__(self, state: dict) -> None:
        if "hvp_damping_coefficient" not in state["state"]:
            logger.warning("Resuming ConjugateGradientOptimizer with lost state.")
        self.state = state["state"]
        self.param_groups = state["param_groups"]

    def unflatten_tensor(self, flattened: Tensor, shapes: List[torch.Size]) -> List[Tensor]:
        """Split a flat vector back into tensors of `shapes` (reference :252-265)."""
        numels = [int(torch.Size(s).numel()) for s in shapes]
        return [
            chunk.view(shape) for chunk, shape in zip(torch.split(flattened, numels), shapes)
        ]
