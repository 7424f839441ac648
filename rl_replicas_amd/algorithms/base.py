"""Shared algorithm infrastructure.

The reference duplicates the epoch loop / metrics / checkpointing
across its five algorithm classes (vpg.py, trpo.py, ppo.py, ddpg.py,
td3.py); here the common machinery lives in one base so the
MI355X-specific plumbing (device residency, one-shot H2D upload,
HIP-graph capture hooks, DP all-reduce) exists in exactly one place.

Public surface per algorithm stays reference-identical:
`learn(...)`, `train(...)`, `save_model(...)` plus the constructor
signatures (see each subclass).  Added beyond the reference:
`load_model(path)` (the reference has no resume path — SURVEY.md §5.4)
and `device` awareness (inferred from the policy parameters).
"""
from __future__ import annotations

import logging
import os
import time
from typing import Dict, List

import numpy as np
import torch

from rl_replicas_amd.metrics_manager import MetricsManager

logger = logging.getLogger(__name__)


class AlgorithmBase:
    policy = None  # set by subclasses

    @property
    def device(self) -> torch.device:
        return next(self.policy.parameters()).device

    # ------------------------------------------------------------------
    # epoch bookkeeping shared by every learn() implementation
    # ------------------------------------------------------------------
    def _begin_learn(self, output_dir: str) -> None:
        self._start_time = time.time()
        self.current_total_steps = 0
        self.current_total_episodes = 0
        os.makedirs(output_dir, exist_ok=True)
        # under data parallelism only rank 0 prints/logs to output_dir;
        # other ranks log quietly to a per-rank subdir
        from rl_replicas_amd.parallel.ddp import get_rank

        rank = get_rank()
        if rank == 0:
            self.metrics_manager = MetricsManager(output_dir)
        else:
            self.metrics_manager = MetricsManager(
                os.path.join(output_dir, f"rank{rank}"), stdout=False
            )

    def _reduce_scalar_mean(self, x):
        """DP hook: mean of a scalar across ranks (identity single-process)."""
        return x

    def _record_sampling_metrics(self, current_epoch: int, episode_returns: List[float], episode_lengths: List[int]) -> None:
        m = self.metrics_manager
        m.record_scalar("epoch", current_epoch)
        m.record_scalar("total_steps", self.current_total_steps)
        m.record_scalar("total_episodes", self.current_total_episodes)
        if len(episode_lengths) > 0:
            m.record_scalar(
                "sampling/average_episode_return",
                float(np.mean(episode_returns)),
                self.current_total_steps,
                tensorboard=True,
            )
            m.record_scalar("sampling/episode_return_std", float(np.std(episode_returns)))
            m.record_scalar("sampling/max_episode_return", float(np.max(episode_returns)))
            m.record_scalar("sampling/min_episode_return", float(np.min(episode_returns)))
            m.record_scalar(
                "sampling/average_episode_length",
                float(np.mean(episode_lengths)),
                self.current_total_steps,
                tensorboard=True,
            )

    def _end_epoch(self, current_epoch: int, model_saving_interval: int, output_dir: str) -> None:
        if self.current_total_steps % model_saving_interval == 0:
            model_path = os.path.join(output_dir, "model.pt")
            logger.debug("Save model")
            self.save_model(current_epoch, model_path)
        self.metrics_manager.record_scalar("time", time.time() - self._start_time)
        self.metrics_manager.dump()

    # ------------------------------------------------------------------
    # checkpointing
    # ------------------------------------------------------------------
    def _checkpoint_dict(self, epoch: int) -> Dict:
        raise NotImplementedError

    def save_model(self, epoch: int, model_path: str) -> None:
        """Write the reference-schema checkpoint dict (model.pt)."""
        torch.save(self._checkpoint_dict(epoch), model_path)

    def load_model(self, model_path: str) -> int:
        """Restore a checkpoint saved by save_model; returns the epoch.

        New relative to the reference (which can save but never load —
        SURVEY.md §5.4): state dict keys follow the same schema, so
        checkpoints written by the reference load here unchanged.
        """
        ckpt = torch.load(model_path, map_location=self.device, weights_only=False)
        self._restore_from_checkpoint(ckpt)
        self.current_total_steps = int(ckpt.get("total_steps", 0))
        return int(ckpt.get("epoch", 0))

    def _restore_from_checkpoint(self, ckpt: Dict) -> None:
        raise NotImplementedError
