"""Replay buffer.

API parity: reference src/rl_replicas/replay_buffer.py:9-74 —
`ReplayBuffer(buffer_size)`, `add_experience(experience)`,
`sample_minibatch(minibatch_size)` returning a dict of numpy arrays
with keys observations/actions/rewards/next_observations/dones, uniform
sampling WITH replacement over the current size.

MI355X redesign: instead of Python-list FIFOs trimmed with `del
list[:n]` (reference :26-49), storage is a pre-allocated ring.  With
`device="cuda"` the ring lives in HBM (288 GB/GPU easily holds the
1e6-transition buffer: HalfCheetah = 1e6 x (17+6+1+17+1) x 4 B ~ 168
MB) and `sample_minibatch_tensors` gathers minibatches entirely
on-device — on-device RNG via torch, gather via the HIP kernel, no
host round-trip on the training hot path (SURVEY.md §2.2 ReplayBuffer
row).  Under data parallelism each rank keeps its own shard (uniform
sampling commutes with sharding — SURVEY.md §2.3 item 3).
"""
from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch

from rl_replicas_amd.experience import Experience


class ReplayBuffer:
    def __init__(self, buffer_size: int = int(1e6), device: Optional[str] = None) -> None:
        self.buffer_size = int(buffer_size)
        self.device = torch.device(device) if device is not None else None
        self.current_size = 0
        self._write = 0
        self._storage: Optional[Dict[str, torch.Tensor]] = None
        # device mirror of current_size, read by the fused gather kernel
        # (lets one captured hipGraph keep sampling as the ring fills)
        self._size_dev: Optional[torch.Tensor] = None
        self._gather_offset = 0

    def _sync_size_dev(self) -> None:
        if self.device is not None and self.device.type == "cuda":
            if self._size_dev is None:
                self._size_dev = torch.zeros(1, dtype=torch.int64, device=self.device)
            self._size_dev.fill_(self.current_size)

    # ------------------------------------------------------------------
    def _allocate(self, obs_dim: int, act_shape) -> None:
        dev = self.device or torch.device("cpu")
        n = self.buffer_size
        self._storage = {
            "observations": torch.empty((n, obs_dim), dtype=torch.float32, device=dev),
            "actions": torch.empty((n, *act_shape), dtype=torch.float32, device=dev),
            "rewards": torch.empty((n,), dtype=torch.float32, device=dev),
            "next_observations": torch.empty((n, obs_dim), dtype=torch.float32, device=dev),
            "dones": torch.empty((n,), dtype=torch.float32, device=dev),
        }

    def add_experience(self, experience: Experience) -> None:
        """Append all transitions of `experience` to the ring."""
        flat = getattr(experience, "_flat_cache", None)
        if (
            flat is not None
            and isinstance(flat.get("observations"), torch.Tensor)
            and "next_observations" in flat
        ):
            # device-resident rollout (DeviceSampler): tensor-to-tensor ring
            # write, no host round trip
            self._add_flat_tensors(flat)
            return
        obs = np.asarray(np.stack(experience.flattened_observations), dtype=np.float32)
        if obs.ndim == 1:
            obs = obs[:, None]
        acts = np.asarray(np.stack(experience.flattened_actions), dtype=np.float32)
        if acts.ndim == 1:
            acts = acts[:, None]
        rews = np.asarray(experience.flattened_rewards, dtype=np.float32)
        next_obs = np.asarray(np.stack(experience.flattened_next_observations), dtype=np.float32)
        if next_obs.ndim == 1:
            next_obs = next_obs[:, None]
        dones = np.asarray(experience.flattened_dones, dtype=np.float32)

        if self._storage is None:
            self._allocate(obs.shape[1], acts.shape[1:])
        assert self._storage is not None

        batch = {
            "observations": obs,
            "actions": acts,
            "rewards": rews,
            "next_observations": next_obs,
            "dones": dones,
        }
        n = len(rews)
        if n > self.buffer_size:
            # keep only the newest buffer_size transitions, like the
            # reference's front-trim (replay_buffer.py:40-49)
            batch = {k: arr[-self.buffer_size :] for k, arr in batch.items()}
            n = self.buffer_size
        dev = self.device or torch.device("cpu")
        pos = self._write
        # ring write, possibly wrapping
        first = min(n, self.buffer_size - pos)
        for key, arr in batch.items():
            t = torch.as_tensor(arr).to(dev, non_blocking=True)
            self._storage[key][pos : pos + first] = t[:first]
            if first < n:
                self._storage[key][: n - first] = t[first:]
        self._write = (pos + n) % self.buffer_size
        self.current_size = min(self.current_size + n, self.buffer_size)
        self._sync_size_dev()

    def _add_flat_tensors(self, flat: Dict[str, torch.Tensor]) -> None:
        batch = {
            "observations": flat["observations"],
            "actions": flat["actions"],
            "rewards": flat["rewards"],
            "next_observations": flat["next_observations"],
            "dones": flat["step_dones"],
        }
        if self._storage is None:
            self._allocate(batch["observations"].shape[1], batch["actions"].shape[1:])
        assert self._storage is not None
        n = batch["rewards"].shape[0]
        if n > self.buffer_size:
            batch = {k: t[-self.buffer_size :] for k, t in batch.items()}
            n = self.buffer_size
        dev = self.device or torch.device("cpu")
        pos = self._write
        first = min(n, self.buffer_size - pos)
        for key, t in batch.items():
            t = t.to(dev, non_blocking=True)
            self._storage[key][pos : pos + first] = t[:first]
            if first < n:
                self._storage[key][: n - first] = t[first:]
        self._write = (pos + n) % self.buffer_size
        self.current_size = min(self.current_size + n, self.buffer_size)
        self._sync_size_dev()

    # ------------------------------------------------------------------
    def sample_minibatch_tensors(self, minibatch_size: int = 32) -> Dict[str, torch.Tensor]:
        """Uniform-with-replacement minibatch as device tensors (hot path)."""
        assert self._storage is not None and self.current_size > 0, "empty replay buffer"
        dev = self.device or torch.device("cpu")
        if dev.type == "cuda":
            idx = torch.randint(0, self.current_size, (minibatch_size,), device=dev)
        else:
            # CPU path keeps the reference's numpy RNG stream semantics
            idx = torch.as_tensor(np.random.randint(0, self.current_size, minibatch_size))
        return {k: v[idx] for k, v in self._storage.items()}

    def sample_minibatch(self, minibatch_size: int = 32) -> Dict[str, np.ndarray]:
        """Reference-compatible numpy minibatch (reference :51-74)."""
        mb = self.sample_minibatch_tensors(minibatch_size)
        return {k: v.cpu().numpy() for k, v in mb.items()}

    # ------------------------------------------------------------------
    def gather_minibatch_fused(
        self,
        minibatch_size: int,
        seed: int,
        offset: int,
        offset_ctr: Optional[torch.Tensor] = None,
    ) -> Dict[str, torch.Tensor]:
        """ONE-kernel minibatch draw+gather from the HBM ring (GPU only).

        Philox-indexed (stateless given seed/offset, graph-replay safe
        via `offset_ctr`); returns the critic input `qin` = [obs|act]
        pre-concatenated alongside the standard keys — replaces
        randint + 5 index_selects + torch.cat on the DDPG/TD3 hot path
        (reference ddpg.py:222-228)."""
        from rl_replicas_amd import ops

        assert self._storage is not None and self.current_size > 0, "empty replay buffer"
        assert self.device is not None and self.device.type == "cuda"
        self._sync_size_dev()
        ext = ops._load_extension()
        st = self._storage
        cap = self.buffer_size
        qin, obs, nxt, rew, dn = ext.replay_gather(
            st["observations"],
            st["actions"].view(cap, -1),
            st["rewards"],
            st["next_observations"],
            st["dones"],
            self._size_dev,
            int(minibatch_size),
            int(seed),
            int(offset),
            offset_ctr,
        )
        return {
            "qin": qin,
            "observations": obs,
            "actions": qin[:, st["observations"].shape[1] :],
            "rewards": rew,
            "next_observations": nxt,
            "dones": dn,
        }

    def __len__(self) -> int:
        return self.current_size
