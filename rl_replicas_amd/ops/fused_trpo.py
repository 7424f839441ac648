"""Analytic Fisher-vector product for TRPO's conjugate-gradient solve.

The reference computes Hv by double backward through the KL graph
(reference: conjugate_gradient_optimizer.py:133-167) — two full
autograd passes per CG iteration.  At TRPO's evaluation point the
policy EQUALS the old policy (the snapshot is synced only after the
update, trpo.py:192), where the KL Hessian is exactly the Fisher
information in its Gauss-Newton form

    H = (1/B) J^T F J     (+ the closed-form log_std / logits blocks),

so Hv factors into a forward-mode JVP through the MLP, a diagonal (or
softmax) scaling, and one fused VJP:

  Gaussian (state-independent log_std, gaussian_policy.py:18-35):
      F_mean = diag(1/sigma^2),  H_logstd = 2 I,  cross terms 0
  Categorical:  F_logits = diag(p) - p p^T

The JVP is a handful of rocBLAS matmuls over the SAVED activations of
ONE fused forward (theta is constant across the whole CG solve, so the
activations are computed once); the VJP is the fused MLP backward
kernel.  ~10 device ops per CG iteration instead of two eager autograd
sweeps.  Validated against the double-backward FVP in
tests/test_gpu_trpo_fvp.py.
"""
from __future__ import annotations

from typing import Callable, List, Optional

import torch
from torch import Tensor

from rl_replicas_amd import ops
from rl_replicas_amd.ops.fused_mlp import ACT_RELU, ACT_TANH, _extract_layers


def _act_grad(code: int, y: Tensor) -> Optional[Tensor]:
    if code == ACT_TANH:
        return 1.0 - y * y
    if code == ACT_RELU:
        return (y > 0).to(y.dtype)
    return None  # identity


def _mlp_jvp(obs: Tensor, weights, hidden, final_out, acts,
             v_ws: List[Tensor], v_bs: List[Tensor]) -> Tensor:
    """Forward-mode tangent through the MLP at saved activations."""
    t: Optional[Tensor] = None
    x = obs
    n = len(weights)
    for l in range(n):
        z = torch.addmm(v_bs[l], x, v_ws[l].t())
        if t is not None:
            z = z + t @ weights[l].t()
        y = final_out if l == n - 1 else hidden[l]
        g = _act_grad(acts[l], y)
        t = z * g if g is not None else z
        x = y
    return t


def make_fvp(policy, obs: Tensor, damping: float) -> Optional[Callable[[Tensor], Tensor]]:
    """Build v -> (H + damping I) v for a Gaussian/Categorical MLP policy,
    or None if the policy/param layout isn't supported (caller falls
    back to the double-backward FVP)."""
    from rl_replicas_amd.ops import fused_onpolicy as fop

    if not (obs.is_cuda and ops.hip_available()):
        return None
    kind = fop._policy_kind(policy)
    if kind is None:
        return None
    mlp = fop._mlp_of(policy)
    if mlp is None:
        return None
    layout = _extract_layers(mlp)
    if layout is None:
        return None
    weights, biases, acts = layout

    # map the optimizer's param order onto (W/b slots, log_std)
    opt_params = [
        p for group in policy.optimizer.param_groups for p in group["params"]
        if p.grad is not None
    ]
    slots: List[tuple] = []  # ("w", l) / ("b", l) / ("log_std",)
    wid = {id(w): l for l, w in enumerate(weights)}
    bid = {id(b): l for l, b in enumerate(biases)}
    log_std = getattr(policy, "log_std", None)
    for p in opt_params:
        if id(p) in wid:
            slots.append(("w", wid[id(p)]))
        elif id(p) in bid:
            slots.append(("b", bid[id(p)]))
        elif log_std is not None and p is log_std:
            slots.append(("log_std",))
        else:
            return None  # unknown parameter in the group
    if kind == "gaussian" and not any(s[0] == "log_std" for s in slots):
        return None

    ext = ops._load_extension()
    obs = obs.contiguous()
    B = obs.shape[0]
    outs = ext.mlp_forward(obs, list(weights), list(biases), acts, True,
                           ops.compute_bf16())
    final_out, hidden = outs[0], list(outs[1:])

    if kind == "gaussian":
        inv_var = torch.exp(-2.0 * log_std.detach())  # [D]
    else:
        probs = torch.softmax(final_out, dim=-1)  # [B, N]

    numels: List[int] = []
    for s in slots:
        if s[0] == "w":
            numels.append(weights[s[1]].numel())
        elif s[0] == "b":
            numels.append(biases[s[1]].numel())
        else:
            numels.append(log_std.numel())

    def fvp(v: Tensor) -> Tensor:
        parts = torch.split(v, numels)
        v_ws: List[Optional[Tensor]] = [None] * len(weights)
        v_bs: List[Optional[Tensor]] = [None] * len(biases)
        v_ls: Optional[Tensor] = None
        for s, chunk in zip(slots, parts):
            if s[0] == "w":
                v_ws[s[1]] = chunk.view(weights[s[1]].shape)
            elif s[0] == "b":
                v_bs[s[1]] = chunk.view(biases[s[1]].shape)
            else:
                v_ls = chunk

        t = _mlp_jvp(obs, weights, hidden, final_out, acts, v_ws, v_bs)
        if kind == "gaussian":
            u = t * inv_var / B
        else:
            u = (t * probs - probs * (t * probs).sum(dim=-1, keepdim=True)) / B
        grads = ext.mlp_backward(u.contiguous(), obs, list(weights), list(biases),
                                 hidden, final_out, acts, ops.compute_bf16())
        n = len(weights)
        dws = grads[1 : 1 + n]
        dbs = grads[1 + n :]
        out_parts = []
        for s in slots:
            if s[0] == "w":
                out_parts.append(dws[s[1]].reshape(-1))
            elif s[0] == "b":
                out_parts.append(dbs[s[1]].reshape(-1))
            else:
                out_parts.append(2.0 * v_ls)
        return torch.cat(out_parts) + damping * v

    return fvp
