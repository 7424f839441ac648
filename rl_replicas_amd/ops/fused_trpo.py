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


class CapturableFVP:
    """Analytic v -> (H + damping I) v with CAPTURE-STABLE buffers.

    The saved activations (and inv_var / probs) live in tensors owned by
    this object; `refresh(obs)` re-runs the fused forward and copies the
    results in.  `__call__` is then a fixed sequence of device kernels
    reading those buffers — so the whole CG solve that calls it can be
    captured ONCE into a hipGraph and replayed every epoch
    (ConjugateGradientOptimizer._CapturedCG), with only the buffer
    refresh running eagerly per epoch.
    """

    def __init__(self, policy, kind: str, obs: Tensor, damping: float,
                 weights, biases, acts, slots: List[tuple], log_std):
        self.kind = kind
        self.damping = damping
        self.weights = weights
        self.biases = biases
        self.acts = acts
        self.slots = slots
        self.log_std = log_std
        self.B = obs.shape[0]
        self.obs = torch.empty_like(obs)
        ext = ops._load_extension()
        outs = ext.mlp_forward(obs.contiguous(), list(weights), list(biases),
                               acts, True, ops.compute_bf16())
        self.final_out = torch.empty_like(outs[0])
        self.hidden = [torch.empty_like(h) for h in outs[1:]]
        if kind == "gaussian":
            self.inv_var = torch.empty_like(log_std.detach())
        else:
            self.probs = torch.empty_like(outs[0])
        self.numels: List[int] = []
        for s in slots:
            if s[0] == "w":
                self.numels.append(weights[s[1]].numel())
            elif s[0] == "b":
                self.numels.append(biases[s[1]].numel())
            else:
                self.numels.append(log_std.numel())
        self.graph_key = (
            kind, tuple(obs.shape), float(damping),
            tuple(id(w) for w in weights) + tuple(id(b) for b in biases),
            ops.compute_bf16(),
        )
        self.refresh(obs)

    @torch.no_grad()
    def refresh(self, obs: Tensor) -> "CapturableFVP":
        ext = ops._load_extension()
        obs = obs.contiguous()
        self.obs.copy_(obs)
        outs = ext.mlp_forward(self.obs, list(self.weights), list(self.biases),
                               self.acts, True, ops.compute_bf16())
        self.final_out.copy_(outs[0])
        for buf, h in zip(self.hidden, outs[1:]):
            buf.copy_(h)
        if self.kind == "gaussian":
            self.inv_var.copy_(torch.exp(-2.0 * self.log_std.detach()))
        else:
            self.probs.copy_(torch.softmax(self.final_out, dim=-1))
        return self

    def __call__(self, v: Tensor) -> Tensor:
        ext = ops._load_extension()
        weights, biases = self.weights, self.biases
        parts = torch.split(v, self.numels)
        v_ws: List[Optional[Tensor]] = [None] * len(weights)
        v_bs: List[Optional[Tensor]] = [None] * len(biases)
        v_ls: Optional[Tensor] = None
        for s, chunk in zip(self.slots, parts):
            if s[0] == "w":
                v_ws[s[1]] = chunk.view(weights[s[1]].shape)
            elif s[0] == "b":
                v_bs[s[1]] = chunk.view(biases[s[1]].shape)
            else:
                v_ls = chunk

        t = _mlp_jvp(self.obs, weights, self.hidden, self.final_out, self.acts,
                     v_ws, v_bs)
        if self.kind == "gaussian":
            u = t * self.inv_var / self.B
        else:
            u = (t * self.probs
                 - self.probs * (t * self.probs).sum(dim=-1, keepdim=True)) / self.B
        grads = ext.mlp_backward(u.contiguous(), self.obs, list(weights),
                                 list(biases), self.hidden, self.final_out,
                                 self.acts, ops.compute_bf16())
        n = len(weights)
        dws = grads[1 : 1 + n]
        dbs = grads[1 + n :]
        out_parts = []
        for s in self.slots:
            if s[0] == "w":
                out_parts.append(dws[s[1]].reshape(-1))
            elif s[0] == "b":
                out_parts.append(dbs[s[1]].reshape(-1))
            else:
                out_parts.append(2.0 * v_ls)
        return torch.cat(out_parts) + self.damping * v


def make_fvp(policy, obs: Tensor, damping: float) -> Optional[Callable[[Tensor], Tensor]]:
    """Build (or refresh the policy-cached) analytic FVP for a
    Gaussian/Categorical MLP policy; None if the policy/param layout
    isn't supported (caller falls back to the double-backward FVP)."""
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

    cached = getattr(policy, "_fvp_cache", None)
    key = (kind, tuple(obs.shape), float(damping),
           tuple(id(w) for w in weights) + tuple(id(b) for b in biases),
           ops.compute_bf16())
    if cached is not None and cached.graph_key == key and cached.slots == slots:
        return cached.refresh(obs)
    fvp = CapturableFVP(policy, kind, obs, damping, weights, biases, acts,
                        slots, log_std)
    policy._fvp_cache = fvp
    return fvp
