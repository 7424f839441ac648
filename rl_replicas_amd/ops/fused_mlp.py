"""Fused MLP execution on CDNA4.

The models in this library are tiny (policy [obs,64,32,act] ~3-4 K
params, off-policy [obs,256,256,act] ~100 K params — BASELINE.md
config) while batches are 4000-row rollouts or 100-row minibatches: the
cost of an eager layer-by-layer forward is pure launch latency (~3.5 us
x 6+ kernels), not FLOPs.  The MI355X-native answer is ONE kernel per
forward: all layer weights are staged into LDS once (13-400 KB fp32 ->
bf16 halves that; larger nets tile), each workgroup owns a row-block of
the batch, and the whole Linear->act->Linear->act->... chain runs from
LDS/registers with MFMA GEMMs and fused activation epilogues
(reference equivalent: nn.Sequential eager, networks/mlp.py:33-41 +
autograd).

Dispatch contract (called from networks.MLP.forward via ops):
  - no_grad context (sampling / evaluation / compute_values / target
    nets — the inference hot path): fused forward kernel.
  - grad-enabled (training): fused forward that stashes activations +
    custom autograd backward running fused dgrad/wgrad kernels.  Not
    re-differentiable: modules that need double backward (TRPO's FVP)
    set `mlp.fused_training = False` to force the eager path for
    grad-enabled forwards.
  - anything unsupported returns NotImplemented -> caller runs eager.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn
from torch import Tensor

# activation codes shared with the HIP kernel (ops/hip/mlp_kernels.hip)
ACT_IDENTITY = 0
ACT_TANH = 1
ACT_RELU = 2

_ACT_CODE = {
    nn.Identity: ACT_IDENTITY,
    nn.Tanh: ACT_TANH,
    nn.ReLU: ACT_RELU,
}

MAX_WIDTH = 256  # widest layer the LDS-staged kernels support


def _extract_layers(mlp) -> Optional[tuple]:
    """Validate the MLP is a supported Linear/act chain; return
    (weights, biases, act_codes) or None."""
    cached = getattr(mlp, "_fused_layout", None)
    if cached is not None:
        return cached if cached != () else None

    mods = list(mlp.network)
    weights: List[Tensor] = []
    biases: List[Tensor] = []
    acts: List[int] = []
    i = 0
    ok = True
    while i < len(mods):
        lin = mods[i]
        if not isinstance(lin, nn.Linear) or lin.bias is None:
            ok = False
            break
        act = mods[i + 1] if i + 1 < len(mods) else nn.Identity()
        code = _ACT_CODE.get(type(act))
        if code is None:
            ok = False
            break
        weights.append(lin.weight)
        biases.append(lin.bias)
        acts.append(code)
        i += 2
    if not ok or not weights:
        mlp._fused_layout = ()
        return None
    if max(w.shape[0] for w in weights) > MAX_WIDTH or weights[0].shape[1] > MAX_WIDTH:
        mlp._fused_layout = ()
        return None
    layout = (weights, biases, acts)
    mlp._fused_layout = layout
    return layout


def try_fused_forward(mlp, input: Tensor):
    from rl_replicas_amd import ops

    if not ops.hip_available():
        return NotImplemented
    if input.dtype != torch.float32:
        return NotImplemented
    layout = _extract_layers(mlp)
    if layout is None:
        return NotImplemented
    weights, biases, acts = layout

    squeeze = input.dim() == 1
    x = input.unsqueeze(0) if squeeze else input
    if x.dim() != 2:
        return NotImplemented

    grad_mode = torch.is_grad_enabled() and (
        x.requires_grad or any(w.requires_grad for w in weights)
    )
    if grad_mode:
        if not getattr(mlp, "fused_training", True):
            return NotImplemented
        out = _FusedMLPTrainFunction.apply(x.contiguous(), acts, *weights, *biases)
    else:
        ext = ops._load_extension()
        out = ext.mlp_forward(
            x.contiguous(), list(weights), list(biases), acts, False, ops.compute_bf16()
        )[0]
    return out.squeeze(0) if squeeze else out


class _FusedMLPTrainFunction(torch.autograd.Function):
    """Forward: one fused kernel (returns output + per-layer pre-activation
    outputs for backward).  Backward: fused dgrad/wgrad kernels.

    NOT double-differentiable — TRPO forces eager (see module docstring).
    """

    @staticmethod
    def forward(ctx, x: Tensor, acts: List[int], *params: Tensor):
        from rl_replicas_amd import ops

        ext = ops._load_extension()
        n = len(params) // 2
        weights = list(params[:n])
        biases = list(params[n:])
        outs = ext.mlp_forward(x, weights, biases, acts, True, ops.compute_bf16())
        # outs = [final_out, act_out_0, ..., act_out_{n-2}]
        ctx.save_for_backward(x, *params, *outs[1:], outs[0])
        ctx.acts = acts
        ctx.n = n
        return outs[0]

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        from rl_replicas_amd import ops

        ext = ops._load_extension()
        saved = ctx.saved_tensors
        n = ctx.n
        x = saved[0]
        weights = list(saved[1 : 1 + n])
        biases = list(saved[1 + n : 1 + 2 * n])
        hidden = list(saved[1 + 2 * n : n * 3])  # n-1 hidden activations
        final_out = saved[-1]
        grads = ext.mlp_backward(
            grad_out.contiguous(), x, weights, biases, hidden, final_out,
            ctx.acts, ops.compute_bf16(),
        )
        # grads = [dx, dW0.., dWn-1, db0.., dbn-1]
        dx = grads[0]
        dws = grads[1 : 1 + n]
        dbs = grads[1 + n :]
        return (dx, None, *dws, *dbs)
