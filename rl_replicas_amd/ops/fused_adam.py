"""Fused multi-tensor Adam for ROCm.

One HIP kernel updates every parameter of the model per step instead
of the per-tensor loop of torch.optim.Adam (reference uses plain
torch.optim.Adam, e.g. benchmarks/run_vpg.py Adam(3e-4)).  State
layout (`exp_avg`, `exp_avg_sq`, `step`) is identical to
torch.optim.Adam so checkpoints interoperate
(`*_optimizer_state_dict` keys in the reference's model.pt schema).
"""
from __future__ import annotations

from typing import List

import torch
from torch import Tensor


class FusedAdam(torch.optim.Adam):
    """torch.optim.Adam with a fused multi-tensor HIP step."""

    @torch.no_grad()
    def step(self, closure=None, *, step_delta: float = 0.0, do_bump: bool = True,
             gate=None):
        """step_delta/do_bump: captured-loop mode — iteration i of a
        hipGraph-captured loop passes step_delta=i, do_bump=False and the
        loop bumps once by num_iters at the end (bump_steps); bitwise
        identical to per-iteration bumps for integer-valued fp32 steps.

        gate: optional 1-elem fp32 device tensor; the update (and bump)
        are skipped while it is 0 — lets a captured loop realize data-
        dependent early stopping (PPO's KL test) without host control
        flow."""
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        from rl_replicas_amd import ops

        ext = ops._load_extension()
        if ext is None:  # pragma: no cover - make_adam gates on hip_available
            return super().step()

        for group in self.param_groups:
            params: List[Tensor] = []
            grads: List[Tensor] = []
            exp_avgs: List[Tensor] = []
            exp_avg_sqs: List[Tensor] = []
            steps: List[Tensor] = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.zeros((), dtype=torch.float32, device=p.device)
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                params.append(p.data)
                grads.append(p.grad.data)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
                steps.append(state["step"])
            if not params:
                continue
            beta1, beta2 = group["betas"]
            ext.fused_adam_(
                params,
                grads,
                exp_avgs,
                exp_avg_sqs,
                steps,
                float(group["lr"]),
                float(beta1),
                float(beta2),
                float(group["eps"]),
                float(group.get("weight_decay", 0.0)),
                float(step_delta),
                bool(do_bump),
                gate,
            )
        return loss

    @torch.no_grad()
    def bump_steps_by(self, amount: Tensor) -> None:
        """Advance every step counter by a DEVICE scalar (the captured
        gated policy loop's executed-iteration count)."""
        from rl_replicas_amd import ops

        ext = ops._load_extension()
        steps = [
            self.state[p]["step"]
            for group in self.param_groups
            for p in group["params"]
            if p in self.state and "step" in self.state[p]
        ]
        if steps:
            ext.adam_bump_dev_(steps, amount)

    @torch.no_grad()
    def bump_steps(self, amount: float) -> None:
        """Advance every step counter by `amount` (one kernel)."""
        from rl_replicas_amd import ops

        ext = ops._load_extension()
        steps = [
            self.state[p]["step"]
            for group in self.param_groups
            for p in group["params"]
            if p in self.state and "step" in self.state[p]
        ]
        if steps:
            ext.adam_bump_(steps, float(amount))


def adam_arg_lists(optimizer, weights, biases):
    """State tensors for the merged reduce+Adam kernel
    (mlp_grad_reduce_adam_f32): returns (m, v, step0, hp) with m/v
    ordered [weights..., biases...], step0 the (shared) pre-bump step
    counter of the first param, and hp the (lr, beta1, beta2, eps,
    weight_decay) of the owning param group.  All listed params must
    live in one group and step in lockstep (true for every optimizer
    this library builds)."""
    import torch as _torch

    params = list(weights) + list(biases)
    group = None
    for g in optimizer.param_groups:
        ids = {id(p) for p in g["params"]}
        if id(params[0]) in ids:
            group = g
            break
    assert group is not None, "params not found in optimizer"
    m, v = [], []
    step0 = None
    for p in params:
        state = optimizer.state[p]
        if len(state) == 0:
            state["step"] = _torch.zeros((), dtype=_torch.float32, device=p.device)
            state["exp_avg"] = _torch.zeros_like(p)
            state["exp_avg_sq"] = _torch.zeros_like(p)
        m.append(state["exp_avg"])
        v.append(state["exp_avg_sq"])
        if step0 is None:
            step0 = state["step"]
    beta1, beta2 = group["betas"]
    hp = (float(group["lr"]), float(beta1), float(beta2), float(group["eps"]),
          float(group.get("weight_decay", 0.0)))
    return m, v, step0, hp
