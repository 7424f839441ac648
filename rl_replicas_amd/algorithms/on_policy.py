"""On-policy algorithm template (VPG / TRPO / PPO).

Reference pipeline (identical in vpg.py:127-192, trpo.py:130-226,
ppo.py:139-223): V(s) forward -> bootstrap -> per-episode scipy scans
-> numpy concat -> torch tensors -> policy update(s) -> 80 value-fn
Adam steps.

MI355X redesign of that pipeline (semantically identical, measured by
the CPU-oracle tests):
  1. the whole ragged rollout is flattened ONCE (Experience.to_flat_batch)
     and uploaded in a single H2D copy;
  2. V(s) for all steps AND all bootstrap observations is ONE batched
     forward (fused MLP kernel on GPU);
  3. discounted returns + GAE for every episode run in ONE segmented
     scan kernel over the episode offsets (ops.gae_advantages_and_returns)
     — no host round-trip, reference quirks preserved (terminated
     episodes still bootstrap deltas with V(s_last); returns use 0);
  4. advantage normalization is a fused mean/std kernel; under data
     parallelism the moments are all-reduced so N-GPU training
     normalizes over the GLOBAL batch exactly like 1-GPU training;
  5. the policy/value update loops run entirely on device; metrics are
     read back once per epoch.
"""
from __future__ import annotations

import logging
from typing import Dict, List

import numpy as np
import torch
import torch.nn.functional as F
from torch import Tensor

from rl_replicas_amd import ops
from rl_replicas_amd.algorithms.base import AlgorithmBase
from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import Policy
from rl_replicas_amd.samplers import Sampler
from rl_replicas_amd.value_function import ValueFunction

logger = logging.getLogger(__name__)


class OnPolicyAlgorithm(AlgorithmBase):
    def __init__(
        self,
        policy: Policy,
        value_function: ValueFunction,
        env,
        sampler: Sampler,
        gamma: float,
        gae_lambda: float,
        num_value_gradients: int,
    ) -> None:
        self.policy = policy
        self.value_function = value_function
        self.env = env
        self.sampler = sampler
        self.gamma = gamma
        self.gae_lambda = gae_lambda
        self.num_value_gradients = num_value_gradients

    # ------------------------------------------------------------------
    def learn(
        self,
        num_epochs: int = 50,
        batch_size: int = 4000,
        model_saving_interval: int = 4000,
        output_dir: str = ".",
    ) -> None:
        """Epoch loop: sample -> metrics -> train -> checkpoint.

        Reference: e.g. ppo.py:72-137.
        """
        import time as _time

        self._begin_learn(output_dir)
        for current_epoch in range(1, num_epochs + 1):
            t0 = _time.perf_counter()
            experience = self.sampler.sample(batch_size, self.policy)
            t1 = _time.perf_counter()

            self.current_total_steps += sum(experience.episode_lengths)
            self.current_total_episodes += sum(experience.episode_dones)

            self._record_sampling_metrics(
                current_epoch, experience.episode_returns, experience.episode_lengths
            )

            self.train(experience)

            # per-phase observability the reference lacks
            self.metrics_manager.record_phase_ms("sample", (t1 - t0) * 1000.0)
            self.metrics_manager.record_phase_ms("train", (_time.perf_counter() - t1) * 1000.0)
            self.metrics_manager.dump_phases(self.current_total_steps)

            self._end_epoch(current_epoch, model_saving_interval, output_dir)
        self.metrics_manager.close()

    # ------------------------------------------------------------------
    def _prepare_batch(self, experience: Experience) -> Dict[str, Tensor]:
        """Device-resident flat batch with advantages and returns.

        On GPU the fp32 payload (obs | actions | rewards | last_obs) is
        packed into one cached pinned buffer and crosses in a single
        H2D copy; offsets/dones are two tiny transfers."""
        device = self.device
        flat = experience.to_flat_batch()
        if isinstance(flat["observations"], Tensor):
            # device-resident rollout (DeviceSampler): nothing to upload
            obs = flat["observations"].to(device)
            actions = flat["actions"].to(device)
            rewards = flat["rewards"].to(device)
            last_obs = flat["last_observations"].to(device)
            return self._finish_batch(flat, obs, actions, rewards, last_obs, device)
        obs_np = flat["observations"]
        act_np = np.asarray(flat["actions"], dtype=np.float32)
        rew_np = flat["rewards"]
        last_np = flat["last_observations"]
        if device.type == "cuda":
            sizes = [a.size for a in (obs_np, act_np, rew_np, last_np)]
            total = int(sum(sizes))
            pinned = getattr(self, "_h2d_pinned", None)
            if pinned is None or pinned.numel() < total:
                pinned = torch.empty(total, dtype=torch.float32, pin_memory=True)
                self._h2d_pinned = pinned
                self._h2d_event = torch.cuda.Event()
            # the previous epoch's non_blocking copy must have drained
            # before the host repacks the same pinned pages (don't rely
            # on implicit sync from later metric readbacks)
            self._h2d_event.synchronize()
            host = pinned[:total].numpy()
            o = 0
            for a in (obs_np, act_np, rew_np, last_np):
                host[o : o + a.size] = a.reshape(-1)
                o += a.size
            dev_flat = pinned[:total].to(device, non_blocking=True)
            self._h2d_event.record()
            s0, s1, s2, s3 = np.cumsum(sizes)
            obs = dev_flat[:s0].view(obs_np.shape)
            actions = dev_flat[s0:s1].view(act_np.shape)
            rewards = dev_flat[s1:s2]
            last_obs = dev_flat[s2:s3].view(last_np.shape)
        else:
            obs = torch.as_tensor(obs_np, dtype=torch.float32, device=device)
            actions = torch.as_tensor(act_np, device=device)
            rewards = torch.as_tensor(rew_np, dtype=torch.float32, device=device)
            last_obs = torch.as_tensor(last_np, dtype=torch.float32, device=device)
        return self._finish_batch(flat, obs, actions, rewards, last_obs, device)

    def _finish_batch(self, flat, obs, actions, rewards, last_obs, device) -> Dict[str, Tensor]:
        offsets = torch.as_tensor(flat["episode_offsets"], device=device)
        dones = torch.as_tensor(flat["episode_dones"], device=device)

        with torch.no_grad():
            values_all = self.value_function(torch.cat([obs, last_obs], dim=0)).flatten()
        num_steps = obs.shape[0]
        values = values_all[:num_steps]
        last_values = values_all[num_steps:]

        advantages, returns = ops.gae_advantages_and_returns(
            rewards, values, last_values, offsets, dones, self.gamma, self.gae_lambda
        )
        # replicate-mode DP: ONE fused all-gather of the post-GAE rows;
        # everything below (normalization, policy/value loops) then runs
        # on the identical global batch on every rank (parallel/ddp.py)
        obs, actions, advantages, returns = self._gather_global_batch(
            obs, actions, advantages, returns
        )
        advantages = self._normalize_advantages(advantages)
        return {
            "observations": obs,
            "actions": actions,
            "advantages": advantages.detach(),
            "discounted_returns": returns.detach(),
        }

    def _gather_global_batch(self, obs, actions, advantages, returns):
        """Hook for replicate-mode DP (identity single-process)."""
        return obs, actions, advantages, returns

    def _normalize_advantages(self, advantages: Tensor) -> Tensor:
        """Global advantage normalization (hook for the DP wrapper)."""
        return ops.normalize(advantages)

    # ------------------------------------------------------------------
    def train(self, experience: Experience) -> None:
        batch = self._prepare_batch(experience)
        obs = batch["observations"]
        actions = batch["actions"]
        advantages = batch["advantages"]
        returns = batch["discounted_returns"]

        policy_metrics = self._update_policy(obs, actions, advantages)

        from rl_replicas_amd.ops import fused_onpolicy

        if fused_onpolicy.value_supported(self, obs):
            value_loss = fused_onpolicy.value_update(self, obs, returns, self.num_value_gradients)
        else:
            value_losses: List[float] = []
            for _ in range(self.num_value_gradients):
                value_losses.append(self.train_value_function(obs, returns).item())
            value_loss = float(np.mean(value_losses))

        m = self.metrics_manager
        for tag, value in policy_metrics.items():
            m.record_scalar(tag, value, self.current_total_steps, tensorboard=True)
        m.record_scalar(
            "value_function/average_loss",
            value_loss,
            self.current_total_steps,
            tensorboard=True,
        )

    # ------------------------------------------------------------------
    def _update_policy(self, obs: Tensor, actions: Tensor, advantages: Tensor) -> Dict[str, float]:
        """Algorithm-specific policy update; returns metric tag->value."""
        raise NotImplementedError

    def _policy_diagnostics(self, obs: Tensor, actions: Tensor) -> Dict[str, float]:
        """entropy / log_prob_std diagnostics (reference logs these
        from the pre-update distribution)."""
        with torch.no_grad():
            dist = self.policy(obs)
            log_probs = dist.log_prob(actions)
            entropies = dist.entropy()
        return {
            "policy/avarage_entropy": float(torch.mean(entropies)),
            "policy/log_prob_std": float(torch.std(log_probs)),
        }

    # ------------------------------------------------------------------
    def train_value_function(self, observations: Tensor, discounted_returns: Tensor) -> Tensor:
        loss = self.compute_value_function_loss(observations, discounted_returns)
        self.value_function.optimizer.zero_grad()
        loss.backward()
        self._all_reduce_gradients(self.value_function)
        self.value_function.optimizer.step()
        return loss.detach()

    def compute_value_function_loss(self, observations: Tensor, discounted_returns: Tensor) -> Tensor:
        values = torch.squeeze(self.value_function(observations), -1)
        return F.mse_loss(values, discounted_returns)

    def _all_reduce_gradients(self, module) -> None:
        """DP hook: no-op single-process; the parallel wrapper overrides."""

    # ------------------------------------------------------------------
    def _checkpoint_dict(self, epoch: int) -> Dict:
        return {
            "epoch": epoch,
            "total_steps": self.current_total_steps,
            "policy_state_dict": self.policy.network.state_dict(),
            "policy_optimizer_state_dict": self.policy.optimizer.state_dict(),
            "value_function_state_dict": self.value_function.network.state_dict(),
            "value_function_optimizer_state_dict": self.value_function.optimizer.state_dict(),
        }

    def _restore_from_checkpoint(self, ckpt: Dict) -> None:
        self.policy.network.load_state_dict(ckpt["policy_state_dict"])
        self.policy.optimizer.load_state_dict(ckpt["policy_optimizer_state_dict"])
        self.value_function.network.load_state_dict(ckpt["value_function_state_dict"])
        self.value_function.optimizer.load_state_dict(ckpt["value_function_optimizer_state_dict"])
