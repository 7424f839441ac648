"""hipGraph-captured device rollout (DeviceSampler fast path).

The eager device rollout issues 3 kernels per env step (fused MLP
forward, Philox action sample, fused env transition) — at 20 steps per
epoch that is ~60 host launches of ~3.5 us each.  Here the WHOLE epoch
rollout (optional lockstep reset / state carry, then steps x
[forward -> sample -> env step], then one RNG-counter bump) is captured
once and replayed as a single hipGraph per epoch.

Randomness across replays: by-value Philox offsets are frozen at
capture, so every RNG kernel adds a device-resident int64 counter
(`offset_ctr`) to its offset; the graph's last node advances the
counter past every offset used, so each replay draws a fresh,
deterministic slice of the stream (same seed => bitwise-identical run,
tests/test_gpu_train.py::test_graphed_rollout_determinism).

Weights/log_std are read by pointer, so the Adam updates between
epochs are visible to the next replay without re-capture.  Re-capture
happens only when the epoch's lockstep-truncation pattern changes
(bounded: the horizon/steps arithmetic cycles through at most two
patterns when horizon % steps == 0, the benchmark shape).

No reference counterpart (the reference samples one env serially on
the host, batch_sampler.py:55-99).
"""
from __future__ import annotations

from typing import List, Tuple

import torch

from rl_replicas_amd import ops
from rl_replicas_amd.ops import fused_onpolicy as fop
from rl_replicas_amd.ops.fused_mlp import _extract_layers


def supported(policy, env) -> bool:
    from rl_replicas_amd.envs.device import DeviceVectorEnv

    if not (env.device.type == "cuda" and ops.hip_available()
            and isinstance(env, DeviceVectorEnv)  # synthetic-dynamics kernels
            and fop._graphs_common(None)
            and fop._policy_kind(policy) == "gaussian"):
        return False
    mlp = fop._mlp_of(policy)
    return mlp is not None and _extract_layers(mlp) is not None


def make_counter(device) -> torch.Tensor:
    """Device RNG counter; base 2^40 keeps the graph streams clear of the
    host-side offsets the eager paths use under the same seeds."""
    return torch.full((1,), 1 << 40, dtype=torch.int64, device=device)


class GraphedRollout:
    """One captured epoch: [reset | state carry] + steps x (mlp_forward ->
    gaussian_sample -> synthetic_env_step) + counter bump."""

    def __init__(self, policy, env, steps: int, cuts: Tuple[int, ...],
                 start_with_reset: bool, ctr: torch.Tensor):
        ext = ops._load_extension()
        N, O, A = env.num_envs, int(env.A.shape[0]), int(env.B.shape[0])
        dev = env.device
        self.env = env
        self.steps = steps
        self.cuts = list(cuts)
        self.start_with_reset = start_with_reset
        # zeros (not empty): capture warmup runs the MLP on these buffers
        # before the first real state lands in the carry slot
        self.obs_full = torch.zeros(steps + 1, N, O, device=dev)
        self.act_buf = torch.zeros(steps, N, A, device=dev)
        self.rew_buf = torch.zeros(steps, N, device=dev)
        self.cut_final = torch.zeros(max(len(cuts), 1), N, O, device=dev)
        # shared across all of a sampler's graphs so truncation-pattern
        # switches never reuse (seed, offset) pairs
        self.ctr = ctr
        policy_seed = torch.initial_seed() & 0x7FFFFFFFFFFFFFFF
        env_seed = env._philox_seed
        weights, biases, acts = _extract_layers(fop._mlp_of(policy))
        log_std = policy.log_std
        cset = {c: j for j, c in enumerate(cuts)}
        compute_bf16 = ops.compute_bf16()

        def body():
            if start_with_reset:
                ext.synthetic_env_reset(N, O, env.A, env_seed, 2 * steps,
                                        self.ctr, self.obs_full[0])
            else:
                self.obs_full[0].copy_(self.obs_full[steps])
            for t in range(steps):
                mean = ext.mlp_forward(self.obs_full[t], list(weights),
                                       list(biases), acts, False, compute_bf16)[0]
                ext.gaussian_sample(mean, log_std.data, policy_seed, t, -1.0,
                                    -1.0, self.ctr, self.act_buf[t])
                do_reset = t in cset
                ext.synthetic_env_step(
                    self.obs_full[t], self.act_buf[t], env.A, env.B, env.w,
                    env.noise, env_seed, 2 * t, do_reset, self.ctr,
                    self.obs_full[t + 1],
                    self.cut_final[cset[t]] if do_reset else self.obs_full[t + 1],
                    self.rew_buf[t],
                )
            # past every offset used: env 2t/2t+1 (t<steps), reset 2*steps,
            # policy t (t<steps)
            ext.counter_add_(self.ctr, 2 * steps + 1)

        # warmup advances ctr (and the state carry buffer); snapshot/restore
        # so the first replay starts exactly where the eager world left off
        state = [self.ctr] if start_with_reset else [self.ctr, self.obs_full]
        self.loop = fop._CapturedLoop(body, state)

    def replay(self) -> None:
        self.loop.replay()

    def final_tensors(self) -> List[torch.Tensor]:
        """Per-cut bootstrap observations, in cut order."""
        return [self.cut_final[j] for j in range(len(self.cuts))]
