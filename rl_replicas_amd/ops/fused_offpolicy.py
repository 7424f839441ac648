"""Fused off-policy (DDPG/TD3) train-step primitives.

The reference's off-policy inner loop (ddpg.py:195-253, td3.py:214-263)
runs 50 minibatch iterations per epoch, each a full torch-autograd
Q-step + delayed actor step.  On GPU this path replaces autograd with
direct kernel calls built from the already-tested primitives:

  gather:       ONE Philox kernel draws indices and gathers the
                minibatch from the HBM ring, emitting the critic input
                [obs|act] pre-concatenated (replay_gather,
                offpolicy_kernels.hip)
  targets:      target-policy fwd -> [TD3] Philox smoothing kernel ->
                target-Q fwd(s) -> fused (min-twin) bootstrap kernel
  q_step:       Q fwd (fused MLP) -> value-MSE loss kernel -> fused MLP
                backward -> fused Adam.
  policy_step:  actor fwd -> Q fwd -> dQ/d(input) via the MLP dgrad
                chain (Q's weight grads are simply not applied,
                mirroring the reference's requires_grad freeze) ->
                actor backward from the action-column slice -> Adam.

No scalar leaves the device inside the loop; the algorithms read the
collected loss tensors once per epoch.

Single-process, the WHOLE `num_train_steps` loop is captured into one
hipGraph (`_GraphedOffPolicy`) and replayed per epoch: iteration i
bakes Adam step_delta=i (one bump at the end, like the captured value
loop in fused_onpolicy.py) and draws its randomness through a device
counter, so replays stay correct as the optimizer state advances and
the ring fills.  Under data parallelism the loop stays eager-fused
(the per-iteration gradient all-reduce cannot live inside a capture).
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch
from torch import Tensor

from rl_replicas_amd import ops
from rl_replicas_amd.ops.fused_mlp import _extract_layers

logger = logging.getLogger(__name__)

# decorrelates this module's Philox streams from the rollout/sampler ones
_SEED_SALT = 0x2545F4914F6CDD1D


def _stream_seed() -> int:
    return (torch.initial_seed() ^ _SEED_SALT) & 0x7FFFFFFFFFFFFFFF


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
    outs = ext.mlp_forward(x, list(weights), list(biases), acts, True,
                           ops.compute_bf16())
    return outs[0], outs[1:], weights, biases, acts


def _backward(mlp, x, grad_out, hidden, final_out, weights, biases, acts):
    ext = ops._load_extension()
    return ext.mlp_backward(
        grad_out.contiguous(), x, list(weights), list(biases), list(hidden),
        final_out, acts, ops.compute_bf16(),
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
           all_reduce_hook, qin: Optional[Tensor] = None,
           step_delta: Optional[float] = None,
           adam=None) -> Tensor:
    """One critic MSE step; returns the loss as a device scalar.

    `qin`: pre-concatenated [obs|act] (the fused gather emits it — skips
    the torch.cat).  `step_delta`: captured-loop mode (deferred Adam
    step-counter bump).  `adam` = (m, v, step0, hp): merged
    reduce+Adam — the backward's reduction applies the update directly
    (captured loop only; all_reduce_hook must be a no-op)."""
    ext = ops._load_extension()
    if qin is None:
        qin = torch.cat([observations, actions], dim=-1).contiguous()
    mlp = q_function.network
    out, hidden, weights, biases, acts = _fwd_saved(mlp, qin)
    dv, scalars = ext.value_mse_loss(out.view(-1), targets.contiguous())
    if adam is not None:
        m, v, step0, hp = adam
        ext.mlp_backward(
            dv.view(out.shape).contiguous(), qin, list(weights), list(biases),
            list(hidden), out, acts, ops.compute_bf16(), m, v, step0, *hp,
            float(step_delta or 0.0),
        )
        return scalars[0]
    grads = _backward(mlp, qin, dv.view(out.shape), hidden, out, weights, biases, acts)
    _apply_grads(q_function, weights, biases, grads)
    all_reduce_hook(q_function)
    if step_delta is None:
        q_function.optimizer.step()
    else:
        q_function.optimizer.step(step_delta=step_delta, do_bump=False)
    return scalars[0]


def policy_step(policy, q_function, observations: Tensor, all_reduce_hook,
                step_delta: Optional[float] = None, adam=None,
                actor_fwd=None) -> Tensor:
    """One deterministic-actor step through a (frozen) critic:
    loss = -mean(Q(s, mu(s)))  (reference ddpg.py:255-273).
    Returns the loss as a device scalar.

    `actor_fwd`: a precomputed _fwd_saved(...) of the actor on
    `observations` — the captured loop runs it on a side stream
    overlapped with the critic steps (it depends only on the policy
    params, which this iteration's critic updates never touch)."""
    ext = ops._load_extension()
    B = observations.shape[0]
    pm = policy.network
    if actor_fwd is None:
        actor_fwd = _fwd_saved(pm, observations.contiguous())
    a_out, a_hidden, a_w, a_b, a_acts = actor_fwd

    qin = torch.cat([observations, a_out], dim=-1).contiguous()
    qm = q_function.network
    q_out, q_hidden, q_w, q_b, q_acts = _fwd_saved(qm, qin)

    # d(-mean(q))/dq = -1/B ; propagate to the Q input, take the action
    # columns, then backprop through the actor.  The critic is frozen
    # (reference semantics), so only the dgrad chain runs — its weight
    # gradients are never computed (input_grad_only).
    dq = torch.full_like(q_out, -1.0 / B)
    q_grads = ext.mlp_backward(
        dq.contiguous(), qin, list(q_w), list(q_b), list(q_hidden), q_out,
        q_acts, ops.compute_bf16(), input_grad_only=True,
    )
    d_qin = q_grads[0]
    d_act = d_qin[:, observations.shape[1] :].contiguous()

    if adam is not None:
        m, v, step0, hp = adam
        ext.mlp_backward(
            d_act, observations, list(a_w), list(a_b), list(a_hidden), a_out,
            a_acts, ops.compute_bf16(), m, v, step0, *hp,
            float(step_delta or 0.0),
        )
        return -q_out.mean()
    a_grads = _backward(pm, observations, d_act, a_hidden, a_out, a_w, a_b, a_acts)
    _apply_grads(policy, a_w, a_b, a_grads)
    all_reduce_hook(policy)
    if step_delta is None:
        policy.optimizer.step()
    else:
        policy.optimizer.step(step_delta=step_delta, do_bump=False)
    return -q_out.mean()


def forward_only(module, x: Tensor) -> Tensor:
    ext = ops._load_extension()
    weights, biases, acts = _extract_layers(module.network)
    return ext.mlp_forward(x.contiguous(), list(weights), list(biases), acts,
                           False, ops.compute_bf16())[0]


# ---------------------------------------------------------------------------
# hipGraph-captured epoch loop
# ---------------------------------------------------------------------------


def _noop_hook(_module) -> None:
    pass


def graph_supported(algo, minibatch_size: int) -> bool:
    """Whole-loop capture needs: GPU ring buffer with storage allocated,
    FusedAdam everywhere, fused-MLP-compatible nets, graphs enabled, and
    no data parallelism (the per-iteration all-reduce stays eager)."""
    import os

    from rl_replicas_amd.ops.fused_adam import FusedAdam
    from rl_replicas_amd.ops.fused_onpolicy import _dp_active

    if os.environ.get("RL_REPLICAS_AMD_DISABLE_GRAPHS", "0") == "1":
        return False
    buf = algo.replay_buffer
    if buf._storage is None or buf.device is None or buf.device.type != "cuda":
        return False
    if buf.current_size <= 0 or not ops.hip_available():
        return False
    if _dp_active(algo):
        return False
    obs0 = buf._storage["observations"][:1]
    modules = [algo.policy, algo.target_policy]
    if hasattr(algo, "q_function"):
        modules += [algo.q_function, algo.target_q_function]
    else:
        modules += [
            algo.q_function_1, algo.q_function_2,
            algo.target_q_function_1, algo.target_q_function_2,
        ]
    for m in modules:
        if not supported(m, obs0):
            return False
    for m in (algo.policy,) + tuple(
        m for m in modules if hasattr(m, "optimizer") and m.optimizer is not None
    ):
        opt = getattr(m, "optimizer", None)
        if opt is not None and not isinstance(opt, FusedAdam):
            return False
    return True


class _GraphedOffPolicy:
    """The whole num_train_steps DDPG/TD3 loop as ONE hipGraph.

    Per iteration (reference td3.py:221-263 order):
      replay_gather -> Q logging fwds -> target chain -> q_step(s)
      [every policy_delay] policy_step + fused Polyak
    RNG: iteration i uses Philox offsets (3i, 3i+1) plus a device
    counter bumped by 3*num_iters inside the graph, so each replay
    draws fresh minibatches/noise.  Adam bumps are deferred (step_delta
    baked per iteration; one bump kernel per optimizer at the end) —
    bitwise-identical to per-iteration stepping.
    """

    def __init__(self, algo, num_iters: int, minibatch_size: int):
        from rl_replicas_amd.ops.fused_onpolicy import (
            _CapturedLoop,
            _ensure_adam_state,
        )
        from rl_replicas_amd.ops.fused_rollout import make_counter

        ext = ops._load_extension()
        buf = algo.replay_buffer
        buf._sync_size_dev()
        dev = buf.device
        st = buf._storage
        cap = buf.buffer_size
        act_flat = st["actions"].view(cap, -1)
        twin = hasattr(algo, "q_function_1")
        delay = getattr(algo, "policy_delay", 1)
        gamma = float(algo.gamma)
        rho = float(algo.polyak_rho)
        seed = _stream_seed()
        self.ctr = make_counter(dev)
        self.num_iters = num_iters
        self.twin = twin

        import numpy as np

        action_limit = float(np.asarray(algo.env.action_space.high).reshape(-1)[0])

        if twin:
            q_fns = [algo.q_function_1, algo.q_function_2]
            tq_fns = [algo.target_q_function_1, algo.target_q_function_2]
        else:
            q_fns = [algo.q_function]
            tq_fns = [algo.target_q_function]

        n_policy = (num_iters + delay - 1) // delay
        from rl_replicas_amd.ops.fused_adam import adam_arg_lists

        q_adams = []
        for q in q_fns:
            qw, qb, _ = _extract_layers(q.network)
            q_adams.append(adam_arg_lists(q.optimizer, qw, qb))
        pw, pb, _ = _extract_layers(algo.policy.network)
        pi_adam = adam_arg_lists(algo.policy.optimizer, pw, pb)
        self.q_losses = [torch.zeros(num_iters, device=dev) for _ in q_fns]
        self.pi_losses = torch.zeros(n_policy, device=dev)
        self.all_q = [
            torch.zeros(num_iters, minibatch_size, device=dev) for _ in q_fns
        ]

        polyak_src = list(algo.policy.network.parameters())
        polyak_dst = list(algo.target_policy.network.parameters())
        for q, tq in zip(q_fns, tq_fns):
            polyak_src += list(q.network.parameters())
            polyak_dst += list(tq.network.parameters())
        polyak_src = [p.data for p in polyak_src]
        polyak_dst = [p.data for p in polyak_dst]

        import os

        use_streams = os.environ.get("RL_REPLICAS_AMD_OFFPOLICY_STREAMS", "1") != "0"
        # side streams for the independent work inside one iteration:
        # the two critics' chains and the target chain have no mutual
        # data dependence until the targets/losses join.  Captured
        # wait_stream edges become graph dependencies, so the replay
        # overlaps these small (16-WG) kernels instead of serializing
        # every launch on one stream.
        s1 = torch.cuda.Stream() if use_streams else None
        s2 = torch.cuda.Stream() if use_streams else None
        s3 = torch.cuda.Stream() if (use_streams and twin) else None

        def body():
            pi_k = 0
            main = torch.cuda.current_stream()
            for i in range(num_iters):
                qin, obs, nxt, rew, dn = ext.replay_gather(
                    st["observations"], act_flat, st["rewards"],
                    st["next_observations"], st["dones"], buf._size_dev,
                    minibatch_size, seed, 3 * i, self.ctr,
                )

                def smoothed_target_input():
                    na = forward_only(algo.target_policy, nxt)
                    if twin:
                        na = ext.td3_smooth(
                            na, seed, 3 * i + 1,
                            float(algo.target_noise_scale),
                            float(algo.target_noise_clip), action_limit,
                            self.ctr,
                        )
                    return torch.cat([nxt, na], dim=-1).contiguous()

                def target_chain():
                    qt_in = smoothed_target_input()
                    if twin:
                        q1t = forward_only(tq_fns[0], qt_in).view(-1)
                        q2t = forward_only(tq_fns[1], qt_in).view(-1)
                        return ext.q_target_min2(rew, dn, q1t, q2t, gamma)
                    qt = forward_only(tq_fns[0], qt_in).view(-1)
                    return ext.q_target(rew, dn, qt, gamma)

                if use_streams:
                    # every fork originates from (and joins back to) the
                    # capture-origin stream — nested forks from a side
                    # stream break hipGraph capture
                    s1.wait_stream(main)
                    s2.wait_stream(main)
                    # main: q1 logging fwd (reference td3.py:230-236)
                    self.all_q[0][i].copy_(forward_only(q_fns[0], qin).view(-1))
                    with torch.cuda.stream(s1):
                        if twin:
                            self.all_q[1][i].copy_(
                                forward_only(q_fns[1], qin).view(-1)
                            )
                    with torch.cuda.stream(s2):
                        qt_in = smoothed_target_input()
                    main.wait_stream(s2)
                    if twin:
                        # q1' on main overlaps q2' on s3
                        s3.wait_stream(main)
                        q1t = forward_only(tq_fns[0], qt_in).view(-1)
                        with torch.cuda.stream(s3):
                            q2t = forward_only(tq_fns[1], qt_in).view(-1)
                        main.wait_stream(s3)
                        targets = ext.q_target_min2(rew, dn, q1t, q2t, gamma)
                    else:
                        qt = forward_only(tq_fns[0], qt_in).view(-1)
                        targets = ext.q_target(rew, dn, qt, gamma)
                    main.wait_stream(s1)
                    # critic steps in parallel: q1 on main, q2 on s1
                    s1.wait_stream(main)
                    actor_fwd = None
                    if i % delay == 0:
                        # prefetch the actor forward for this iteration's
                        # policy step: depends only on policy params +
                        # obs, both untouched by the critic steps
                        s2.wait_stream(main)
                        with torch.cuda.stream(s2):
                            actor_fwd = _fwd_saved(
                                algo.policy.network, obs.contiguous()
                            )
                    self.q_losses[0][i].copy_(
                        q_step(q_fns[0], obs, None, targets, _noop_hook,
                               qin=qin, step_delta=float(i), adam=q_adams[0])
                    )
                    if twin:
                        with torch.cuda.stream(s1):
                            self.q_losses[1][i].copy_(
                                q_step(q_fns[1], obs, None, targets, _noop_hook,
                                       qin=qin, step_delta=float(i),
                                       adam=q_adams[1])
                            )
                    main.wait_stream(s1)
                    if actor_fwd is not None:
                        main.wait_stream(s2)
                else:
                    for q, buf_q in zip(q_fns, self.all_q):
                        buf_q[i].copy_(forward_only(q, qin).view(-1))
                    targets = target_chain()
                    for q, buf_l, qa in zip(q_fns, self.q_losses, q_adams):
                        buf_l[i].copy_(
                            q_step(q, obs, None, targets, _noop_hook, qin=qin,
                                   step_delta=float(i), adam=qa)
                        )
                if i % delay == 0:
                    self.pi_losses[pi_k].copy_(
                        policy_step(algo.policy, q_fns[0], obs, _noop_hook,
                                    step_delta=float(pi_k), adam=pi_adam,
                                    actor_fwd=(actor_fwd if use_streams else None))
                    )
                    ext.fused_polyak_(polyak_src, polyak_dst, rho)
                    pi_k += 1
            for q in q_fns:
                q.optimizer.bump_steps(float(num_iters))
            algo.policy.optimizer.bump_steps(float(pi_k))
            ext.counter_add_(self.ctr, 3 * num_iters)

        state = [t for m in ([algo.policy, algo.target_policy] + q_fns + tq_fns)
                 for t in (p.data for p in m.network.parameters())]
        state += _ensure_adam_state(algo.policy.optimizer)
        for q in q_fns:
            state += _ensure_adam_state(q.optimizer)
        state.append(self.ctr)
        self.loop = _CapturedLoop(body, state)

    def run(self) -> Dict[str, float]:
        self.loop.replay()
        metrics: Dict[str, float] = {
            "policy/average_loss": float(self.pi_losses.mean()),
        }
        names = ["q-function_1", "q-function_2"] if self.twin else ["q-function"]
        for name, losses, qv in zip(names, self.q_losses, self.all_q):
            metrics[f"{name}/average_loss"] = float(losses.mean())
            metrics[f"{name}/avarage_q-value"] = float(qv.mean())
            metrics[f"{name}/max_q-value"] = float(qv.max())
            metrics[f"{name}/min_q-value"] = float(qv.min())
        return metrics


def graphed_epoch(algo, num_train_steps: int, minibatch_size: int) -> Dict[str, float]:
    """Run (capturing on first use) the graphed off-policy epoch."""
    from rl_replicas_amd.ops.fused_onpolicy import _get_cached_graph

    key = (
        num_train_steps,
        minibatch_size,
        id(algo.replay_buffer),
        tuple(id(p) for p in algo.policy.network.parameters()),
        ops.compute_bf16(),
    )
    graphed = _get_cached_graph(
        algo, "_offpolicy_graph", key,
        lambda: _GraphedOffPolicy(algo, num_train_steps, minibatch_size),
    )
    return graphed.run()
