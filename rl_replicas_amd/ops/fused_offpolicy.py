"""Fused off-policy (DDPG/TD3) train-step primitives.

The reference's off-policy inner loop (ddpg.py:195-253, td3.py:214-263)
runs 50 minibatch iterations per epoch, each a full torch-autograd
Q-step + delayed actor step.  On GPU this path replaces autograd with
direct kernel calls built from the already-tested primitives:

  q_step:       Q fwd (fused MLP over cat[obs,act]) -> value-MSE loss
                kernel -> fused MLP backward -> fused Adam.
  policy_step:  actor fwd -> Q fwd -> dQ/d(input) via the MLP dgrad
                chain (Q's weight grads are simply not applied,
                mirroring the reference's requires_grad freeze) ->
                actor backward from the action-column slice -> Adam.

No scalar leaves the device inside the loop; the algorithms read the
collected loss tensors once per epoch.
"""
from __future__ import annotations



import torch
from torch import Tensor

from rl_replicas_amd import ops
from rl_replicas_amd.ops.fused_mlp import _extract_layers


def supported(module, obs: Tensor) -> bool:
    """module: QFunction / DeterministicPolicy with an MLP network."""
    from rl_replicas_amd.networks import MLP

    return (
        obs.is_cuda
        and ops.hip_available()
        and isinstance(getattr(module, "network", None), MLP)
        and _extract_layers(module.network) is not None
    )


def _fwd_saved(mlp, x: Tensor):
    ext = ops._load_extension()
    weights, biases, acts = _extract_layers(mlp)
    outs = ext.mlp_forward(x, list(weights), list(biases), acts, True)
    return outs[0], outs[1:], weights, biases, acts


def _backward(mlp, x, grad_out, hidden, final_out, weights, biases, acts):
    ext = ops._load_extension()
    return ext.mlp_backward(
        grad_out.contiguous(), x, list(weights), list(biases), list(hidden),
        final_out, acts,
    )


def _apply_grads(module, weights, biases, grads, extra=()):
    n = len(weights)
    for w, dw in zip(weights, grads[1 : 1 + n]):
        w.grad = dw
    for b, db in zip(biases, grads[1 + n :]):
        b.grad = db
    for p, g in extra:
        p.grad = g


def q_step(q_function, observations: Tensor, actions: Tensor, targets: Tensor,
           all_reduce_hook) -> Tensor:
    """One critic MSE step; returns the loss as a device scalar."""
    ext = ops._load_extension()
    qin = torch.cat([observations, actions], dim=-1).contiguous()
    mlp = q_function.network
    out, hidden, weights, biases, acts = _fwd_saved(mlp, qin)
    dv, scalars = ext.value_mse_loss(out.view(-1), targets.contiguous())
    grads = _backward(mlp, qin, dv.view(out.shape), hidden, out, weights, biases, acts)
    _apply_grads(q_function, weights, biases, grads)
    all_reduce_hook(q_function)
    q_function.optimizer.step()
    return scalars[0]


def policy_step(policy, q_function, observations: Tensor, all_reduce_hook) -> Tensor:
    """One deterministic-actor step through a (frozen) critic:
    loss = -mean(Q(s, mu(s)))  (reference ddpg.py:255-273).
    Returns the loss as a device scalar."""
    ext = ops._load_extension()
    B = observations.shape[0]
    pm = policy.network
    a_out, a_hidden, a_w, a_b, a_acts = _fwd_saved(pm, observations.contiguous())

    qin = torch.cat([observations, a_out], dim=-1).contiguous()
    qm = q_function.network
    q_out, q_hidden, q_w, q_b, q_acts = _fwd_saved(qm, qin)

    # d(-mean(q))/dq = -1/B ; propagate to the Q input, take the action
    # columns, then backprop through the actor.  Q's weight grads are
    # computed but never applied (the reference freezes the critic).
    dq = torch.full_like(q_out, -1.0 / B)
    q_grads = _backward(qm, qin, dq, q_hidden, q_out, q_w, q_b, q_acts)
    d_qin = q_grads[0]
    d_act = d_qin[:, observations.shape[1] :].contiguous()

    a_grads = _backward(pm, observations, d_act, a_hidden, a_out, a_w, a_b, a_acts)
    _apply_grads(policy, a_w, a_b, a_grads)
    all_reduce_hook(policy)
    policy.optimizer.step()
    return -q_out.mean()


def forward_only(module, x: Tensor) -> Tensor:
    ext = ops._load_extension()
    weights, biases, acts = _extract_layers(module.network)
    return ext.mlp_forward(x.contiguous(), list(weights), list(biases), acts, False)[0]
