"""Data parallelism over RCCL/xGMI.

The reference has no distributed anything.  This
module adds single-node data parallelism designed for the MI355X xGMI
topology: one process per GPU (`torch.distributed`, backend "nccl" =
RCCL on ROCm), each rank sampling its own env shard and all-reducing
gradients.

Design choices (xGMI-first):
* Gradients here are KILOBYTES (policy+value ~4 K params fp32), so
  collectives are pure latency.  We therefore send ONE fused flat
  buffer per optimizer step (torch._utils flatten/unflatten) rather
  than per-parameter or bucketed messages; overlap machinery would
  only add launch latency at this size.
* Advantage normalization must be GLOBAL to reproduce single-GPU
  numerics: ranks all-reduce [sum, sum_sq, count] (one 3-float
  message) and normalize with the global Bessel-corrected std.
* PPO's early-stop KL is all-reduced (mean) so every rank takes the
  same number of policy steps — replicas stay bitwise identical.
* Rank failure aborts the job (no elasticity on one node; the
  reference's only robustness is numerical — SURVEY.md §5.3).

Works on CPU with the gloo backend for testing (tests/test_parallel_cpu.py).
"""
from __future__ import annotations

import logging
import os
from typing import List, Optional

import torch
import torch.distributed as dist
from torch import Tensor
from torch._utils import _flatten_dense_tensors, _unflatten_dense_tensors

logger = logging.getLogger(__name__)


def distributed_is_active() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if distributed_is_active() else 0


def get_world_size() -> int:
    return dist.get_world_size() if distributed_is_active() else 1


def init_from_env(backend: Optional[str] = None, timeout_s: float = 600.0) -> int:
    """Initialize torch.distributed from torchrun env vars; returns rank.

    Selects RCCL ("nccl") when a GPU is visible, gloo otherwise, binds
    the process to its LOCAL_RANK GPU, and is a no-op outside a
    distributed launch (WORLD_SIZE unset or 1 with no MASTER_ADDR).
    `timeout_s` bounds every collective: a dead rank aborts the whole
    job after that long (single-node DP has no elasticity by design —
    SURVEY.md §5.3).
    """
    if distributed_is_active():
        return dist.get_rank()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1 and "MASTER_ADDR" not in os.environ:
        return 0
    if backend is None:
        backend = os.environ.get("RL_REPLICAS_AMD_DIST_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo"
        )
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if torch.cuda.is_available():
        # modulo lets multi-rank smoke tests share one GPU (gloo backend)
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
    from datetime import timedelta

    dist.init_process_group(backend=backend, timeout=timedelta(seconds=timeout_s))
    logger.info(
        "initialized process group: backend=%s rank=%d world=%d",
        backend,
        dist.get_rank(),
        dist.get_world_size(),
    )
    return dist.get_rank()


# ---------------------------------------------------------------------------
def _broadcast_module(module: torch.nn.Module, src: int = 0) -> None:
    for t in list(module.parameters()) + list(module.buffers()):
        dist.broadcast(t.data, src=src)


def all_reduce_gradients(module: torch.nn.Module) -> None:
    """Average gradients across ranks as ONE fused flat message."""
    if not distributed_is_active():
        return
    grads: List[Tensor] = [p.grad for p in module.parameters() if p.grad is not None]
    if not grads:
        return
    flat = _flatten_dense_tensors(grads)
    dist.all_reduce(flat)
    flat.div_(dist.get_world_size())
    for g, synced in zip(grads, _unflatten_dense_tensors(flat, grads)):
        g.copy_(synced)


def global_normalize(x: Tensor) -> Tensor:
    """(x - mean)/std over the GLOBAL (all-rank) batch, Bessel-corrected."""
    if not distributed_is_active():
        from rl_replicas_amd import ops

        return ops.normalize(x)
    stats = torch.stack([x.sum(), (x * x).sum(), torch.tensor(float(x.numel()), device=x.device, dtype=x.dtype)])
    dist.all_reduce(stats)
    total, total_sq, count = stats[0], stats[1], stats[2]
    mean = total / count
    var = (total_sq - count * mean * mean) / (count - 1)
    return (x - mean) / torch.sqrt(var)


def all_reduce_mean_scalar(x: Tensor) -> Tensor:
    if not distributed_is_active():
        return x
    x = x.clone()
    dist.all_reduce(x)
    return x / dist.get_world_size()


def gather_global_batch(obs: Tensor, actions: Tensor, advantages: Tensor,
                        returns: Tensor):
    """ONE fused all-gather of the post-GAE flat batch (replicate-mode
    data parallelism).

    The on-policy models here are kilobyte-scale, so per-iteration
    gradient all-reduces are pure xGMI latency (~240 small collectives
    per epoch).  Instead every rank gathers all ranks' (obs | actions |
    advantages | returns) rows — ~400 KB per rank at the reference
    config, ONE collective per epoch — and then runs the IDENTICAL
    global-batch update locally: zero further communication, replicas
    bitwise-identical by determinism of the kernels, and the update
    math equals the single-GPU global-batch math by construction.
    """
    if not distributed_is_active():
        return obs, actions, advantages, returns
    world = dist.get_world_size()
    n = obs.shape[0]
    act2 = actions.reshape(n, -1)
    if actions.dtype == obs.dtype:
        payload = torch.cat(
            [obs, act2.to(obs.dtype), advantages.view(n, 1), returns.view(n, 1)],
            dim=1,
        ).contiguous()
        chunks = [torch.empty_like(payload) for _ in range(world)]
        dist.all_gather(chunks, payload)
        out = torch.cat(chunks, dim=0)
        O = obs.shape[1]
        A = act2.shape[1]
        g_obs = out[:, :O].contiguous()
        g_act = out[:, O : O + A].contiguous().reshape(world * n, *actions.shape[1:])
        g_adv = out[:, O + A].contiguous()
        g_ret = out[:, O + A + 1].contiguous()
        return g_obs, g_act, g_adv, g_ret
    # mixed dtypes (e.g. integer actions): per-tensor gathers
    gathered = []
    for t in (obs, actions, advantages, returns):
        chunks = [torch.empty_like(t) for _ in range(world)]
        dist.all_gather(chunks, t.contiguous())
        gathered.append(torch.cat(chunks, dim=0))
    return tuple(gathered)


# ---------------------------------------------------------------------------
def enable_data_parallel(algorithm, mode: Optional[str] = None) -> None:
    """Wire an algorithm instance for multi-rank training.

    Broadcasts every module's initial parameters from rank 0 and
    installs the DP hooks the algorithm templates call.  The caller is
    responsible for per-rank sampler seeds (utils.set_seed_for_rank)
    and for scaling batch sizes.

    Two on-policy modes (RL_REPLICAS_AMD_DP_MODE or `mode`):

    * "replicate" (default): ONE fused all-gather of the post-GAE flat
      batch per epoch (`gather_global_batch`); every rank then runs the
      identical global-batch update with ZERO per-iteration collectives
      — the right shape for kilobyte-scale models on xGMI, where
      per-step all-reduces are pure latency.  Replicas stay bitwise
      identical because the update inputs are identical and the kernels
      deterministic.
    * "allreduce": the classic per-iteration fused-flat gradient
      all-reduce + global advantage moments + KL-synced early stop.

    Off-policy (DDPG/TD3) always uses the all-reduce hooks: replay
    buffers are sharded per rank, so minibatch
    gradients genuinely differ per rank and are averaged per step.
    """
    if not distributed_is_active():
        logger.warning("enable_data_parallel called without an initialized process group; no-op")
        return

    import types

    import torch.nn as nn

    if mode is None:
        mode = os.environ.get("RL_REPLICAS_AMD_DP_MODE", "replicate")
    assert mode in ("replicate", "allreduce"), mode
    from rl_replicas_amd.algorithms.on_policy import OnPolicyAlgorithm

    replicate = mode == "replicate" and isinstance(algorithm, OnPolicyAlgorithm)

    for name in (
        "policy",
        "old_policy",
        "value_function",
        "q_function",
        "q_function_1",
        "q_function_2",
        "target_policy",
        "target_q_function",
        "target_q_function_1",
        "target_q_function_2",
    ):
        module = getattr(algorithm, name, None)
        if isinstance(module, nn.Module):
            _broadcast_module(module)

    if replicate:
        def _all_reduce_gradients(self, module) -> None:
            pass  # the whole global batch is local: grads are already global

        def _normalize_advantages(self, advantages: Tensor) -> Tensor:
            from rl_replicas_amd import ops

            return ops.normalize(advantages)  # input is the gathered vector

        def _reduce_scalar_mean(self, x: Tensor) -> Tensor:
            return x  # computed on the identical global batch everywhere

        def _gather_global_batch(self, obs, actions, advantages, returns):
            return gather_global_batch(obs, actions, advantages, returns)

        algorithm._gather_global_batch = types.MethodType(_gather_global_batch, algorithm)
        algorithm._dp_replicate = True
    else:
        def _all_reduce_gradients(self, module) -> None:
            all_reduce_gradients(module)

        def _normalize_advantages(self, advantages: Tensor) -> Tensor:
            return global_normalize(advantages)

        def _reduce_scalar_mean(self, x: Tensor) -> Tensor:
            return all_reduce_mean_scalar(x)

    algorithm._all_reduce_gradients = types.MethodType(_all_reduce_gradients, algorithm)
    algorithm._normalize_advantages = types.MethodType(_normalize_advantages, algorithm)
    algorithm._reduce_scalar_mean = types.MethodType(_reduce_scalar_mean, algorithm)
    algorithm._dp_enabled = True
