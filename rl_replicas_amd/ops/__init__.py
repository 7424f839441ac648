"""Compute ops: HIP/CDNA4 kernels with PyTorch reference fallbacks.

Every hot primitive of the training pipelines has two
implementations:

* a hand-written CDNA4 HIP kernel (`rl_replicas_amd/_hip_ops*.so`,
  sources under `rl_replicas_amd/ops/hip/`, compiled for gfx950) used
  whenever the tensors live on a ROCm GPU, and
* a plain-PyTorch fp32 implementation that is (a) the CPU path and
  (b) the numerics oracle the kernels are tested against
  (tests/test_gpu_ops.py).

Loud-failure contract: on a GPU box the HIP extension MUST load — if a
CUDA tensor reaches an op and the extension is missing, we raise
instead of silently falling back to eager PyTorch.  Set
``RL_REPLICAS_AMD_ALLOW_EAGER=1`` to override (debug only).
"""
from __future__ import annotations

import logging
import os
from typing import List, Optional, Tuple

import torch
from torch import Tensor

logger = logging.getLogger(__name__)

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import importlib

        _EXT = importlib.import_module("rl_replicas_amd._hip_ops")
        logger.info("HIP ops extension loaded: %s", _EXT.__file__)
    except Exception as e:  # pragma: no cover - exercised only on GPU boxes
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def hip_available() -> bool:
    return _load_extension() is not None


def _allow_eager() -> bool:
    return os.environ.get("RL_REPLICAS_AMD_ALLOW_EAGER", "0") == "1"


def compute_bf16() -> int:
    """1 when the MLP GEMM compute dtype is bf16 (MFMA
    v_mfma_f32_16x16x32_bf16, fp32 accumulate) instead of exact fp32
    MFMA.  Set RL_REPLICAS_AMD_COMPUTE_DTYPE=bf16 (or call
    set_compute_dtype) before building models; loss/scan/optimizer
    math stays fp32 (mixed precision)."""
    return 1 if os.environ.get("RL_REPLICAS_AMD_COMPUTE_DTYPE", "fp32") == "bf16" else 0


def set_compute_dtype(dtype: str) -> None:
    assert dtype in ("fp32", "bf16"), dtype
    os.environ["RL_REPLICAS_AMD_COMPUTE_DTYPE"] = dtype


def wants_hip(t) -> bool:
    """True if `t` (tensor/parameter) is on GPU and the HIP path should run.

    Raises if on GPU without the extension (unless eager override) —
    GPU runs must never silently use the fallback.
    """
    tensor = t.data if hasattr(t, "data") else t
    if not isinstance(tensor, Tensor) or not tensor.is_cuda:
        return False
    if hip_available():
        return True
    if _allow_eager():
        return False
    raise RuntimeError(
        "rl_replicas_amd: tensor is on GPU but the HIP ops extension failed to "
        f"load ({_EXT_ERR}). Build it with `python setup.py build_ext --inplace` "
        "(PYTORCH_ROCM_ARCH=gfx950) or set RL_REPLICAS_AMD_ALLOW_EAGER=1 to "
        "run eager PyTorch (debug only)."
    )


# ---------------------------------------------------------------------------
# normalize: (x - mean) / std  (Bessel-corrected std, matching torch.std)
# ---------------------------------------------------------------------------
def normalize(x: Tensor) -> Tensor:
    if wants_hip(x) and x.dtype == torch.float32 and x.dim() == 1:
        return _HipNormalize.apply(x.contiguous())
    return (x - torch.mean(x)) / torch.std(x)


class _HipNormalize(torch.autograd.Function):
    """y = (x - mu) / sigma with sigma treated as constant for backward.

    The advantage tensor is always detached in the algorithms (GAE is
    computed under no_grad), so the backward here only needs to support
    the trivial pass-through case; we still implement the exact
    derivative wrt x treating mu/sigma as functions of x is NOT needed
    because advantages never require grad.
    """

    @staticmethod
    def forward(ctx, x: Tensor) -> Tensor:
        ext = _load_extension()
        return ext.normalize(x)

    @staticmethod
    def backward(ctx, grad: Tensor):  # pragma: no cover - never needed
        raise RuntimeError("normalize() backward is not used (inputs are detached)")


# ---------------------------------------------------------------------------
# fused Polyak: target <- rho*target + (1-rho)*param over a tensor list
# ---------------------------------------------------------------------------
def fused_polyak_(params: List[Tensor], targets: List[Tensor], rho: float) -> None:
    if targets and wants_hip(targets[0]):
        ext = _load_extension()
        ext.fused_polyak_(params, targets, float(rho))
        return
    torch._foreach_mul_(targets, rho)
    torch._foreach_add_(targets, params, alpha=1.0 - rho)


# ---------------------------------------------------------------------------
# segmented GAE + discounted returns over a ragged rollout
# ---------------------------------------------------------------------------
def gae_advantages_and_returns(
    rewards: Tensor,
    values: Tensor,
    last_values: Tensor,
    episode_offsets: Tensor,
    episode_dones: Tensor,
    gamma: float,
    gae_lambda: float,
) -> Tuple[Tensor, Tensor]:
    """Whole-rollout GAE + lambda-free discounted returns in one pass.

    Exact semantics of the reference pipeline (ppo.py:140-161):
      boot_e   = 0 if episode_dones[e] else last_values[e]
      returns: acc=boot_e; t=L-1..0: acc = r[t] + gamma*acc
      deltas:  d[t] = r[t] + gamma*v[t+1] - v[t]  (v[L] := last_values[e],
               NOT zeroed on termination — the reference quirk, kept)
      advs:    acc=0; t=L-1..0: acc = d[t] + gamma*lambda*acc

    Shapes: rewards/values [T], last_values/episode_dones [N],
    episode_offsets [N+1] (int32/64).  Returns (advantages[T], returns[T]).
    """
    if wants_hip(rewards):
        ext = _load_extension()
        return ext.segmented_gae(
            rewards.contiguous(),
            values.contiguous(),
            last_values.contiguous(),
            episode_offsets.to(torch.int32).contiguous(),
            episode_dones.to(torch.int32).contiguous(),
            float(gamma),
            float(gae_lambda),
        )
    return _gae_reference(
        rewards, values, last_values, episode_offsets, episode_dones, gamma, gae_lambda
    )


def _gae_reference(rewards, values, last_values, episode_offsets, episode_dones, gamma, gae_lambda):
    """CPU oracle: per-episode backward recurrences in numpy (identical
    math to the kernel; plain float arithmetic instead of per-element
    torch dispatch keeps the CPU path usable at 4000-row batches)."""
    import numpy as np

    r = rewards.detach().cpu().numpy().astype(np.float32)
    v = values.detach().cpu().numpy().astype(np.float32)
    lv = last_values.detach().cpu().numpy().astype(np.float32)
    offs = episode_offsets.tolist()
    dones = [bool(d) for d in episode_dones]
    adv = np.empty_like(r)
    ret = np.empty_like(r)
    gl = np.float32(gamma * gae_lambda)
    g = np.float32(gamma)
    for e in range(len(offs) - 1):
        lo, hi = offs[e], offs[e + 1]
        last_v = lv[e]
        ret_acc = np.float32(0.0) if dones[e] else last_v
        adv_acc = np.float32(0.0)
        for t in range(hi - 1, lo - 1, -1):
            ret_acc = r[t] + g * ret_acc
            ret[t] = ret_acc
            v_next = v[t + 1] if t + 1 < hi else last_v
            delta = r[t] + g * v_next - v[t]
            adv_acc = delta + gl * adv_acc
            adv[t] = adv_acc
    device = rewards.device
    return (
        torch.as_tensor(adv, dtype=rewards.dtype, device=device),
        torch.as_tensor(ret, dtype=rewards.dtype, device=device),
    )


# ---------------------------------------------------------------------------
# fused MLP forward (dispatched from networks.MLP)
# ---------------------------------------------------------------------------
def mlp_fused_forward(mlp, input: Tensor):
    """Run the whole MLP in fused HIP kernels; NotImplemented if the
    architecture isn't supported (falls back to nn.Sequential)."""
    from rl_replicas_amd.ops.fused_mlp import try_fused_forward

    return try_fused_forward(mlp, input)


# ---------------------------------------------------------------------------
# fused Adam (multi-tensor)
# ---------------------------------------------------------------------------
def make_adam(params, lr: float, **kwargs):
    """Adam factory: fused multi-tensor HIP Adam on GPU, torch.optim.Adam
    otherwise.  State-dict compatible with torch.optim.Adam."""
    from rl_replicas_amd.ops.fused_adam import FusedAdam

    params = list(params)
    p0 = params[0] if params else None
    dev_param = p0["params"][0] if isinstance(p0, dict) else p0
    if dev_param is not None and isinstance(dev_param, Tensor) and dev_param.is_cuda and hip_available():
        return FusedAdam(params, lr=lr, **kwargs)
    return torch.optim.Adam(params, lr=lr, **kwargs)


# ---------------------------------------------------------------------------
# fused DDPG/TD3 target:  y = r + gamma*(1-d)*q_next
# ---------------------------------------------------------------------------
def q_target(rewards: Tensor, dones: Tensor, q_next: Tensor, gamma: float) -> Tensor:
    if wants_hip(rewards):
        ext = _load_extension()
        return ext.q_target(rewards.contiguous(), dones.contiguous(), q_next.contiguous(), float(gamma))
    return rewards + gamma * (1.0 - dones) * q_next


def q_target_min2(rewards: Tensor, dones: Tensor, q1: Tensor, q2: Tensor, gamma: float) -> Tensor:
    """TD3 min-twin bootstrap target r + gamma*(1-d)*min(q1,q2)
    (reference td3.py:335-341) in one kernel."""
    if wants_hip(rewards):
        ext = _load_extension()
        return ext.q_target_min2(
            rewards.contiguous(), dones.contiguous(), q1.contiguous(),
            q2.contiguous(), float(gamma)
        )
    return rewards + gamma * (1.0 - dones) * torch.min(q1, q2)


def td3_smooth(actions: Tensor, scale: float, clip: float, limit: float,
               seed: int, offset: int, offset_ctr: Optional[Tensor] = None) -> Tensor:
    """Target-policy smoothing clamp(a + clamp(scale*z, +-clip), +-limit)
    (reference td3.py:325-341).  GPU: ONE Philox kernel (graph-replay
    safe via offset_ctr).  CPU: the reference's torch.randn chain, so the
    CPU path keeps the reference's RNG stream semantics."""
    if wants_hip(actions):
        ext = _load_extension()
        return ext.td3_smooth(
            actions.contiguous(), int(seed), int(offset), float(scale),
            float(clip), float(limit), offset_ctr
        )
    epsilon = scale * torch.randn_like(actions)
    epsilon = torch.clamp(epsilon, -clip, clip)
    return torch.clamp(actions + epsilon, -limit, limit)
