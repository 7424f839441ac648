"""Rollout container.

API parity with the reference's `Experience` (reference:
src/rl_replicas/experience.py:6-84): nested per-episode lists
(`observations/actions/rewards/dones` of shape (N episodes, L steps)),
`last_observations`, `episode_returns`, `episode_lengths`, plus the
derived views (`observations_with_last_observation`,
`next_observations`, `episode_dones`, `flattened_*`).

MI355X addition: `to_flat_batch()` materializes the whole rollout as
contiguous numpy arrays + episode offset indices in ONE pass — the
device pipeline uploads these with a single H2D copy per epoch and runs
the segmented GAE scan over the offsets on the GPU (SURVEY.md §2.2
"Experience flatten/concat").
"""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np


class Experience:
    """N episodes of (observation, action, reward, done) steps.

    Shapes: observations (N, L, O*), actions (N, L, A*), rewards (N, L),
    dones (N, L), last_observations (N, O*), episode_returns (N),
    episode_lengths (N).  L varies per episode (ragged).
    """

    def __init__(
        self,
        observations: Optional[List[List[np.ndarray]]] = None,
        actions: Optional[List[List[np.ndarray]]] = None,
        rewards: Optional[List[List[float]]] = None,
        last_observations: Optional[List[np.ndarray]] = None,
        dones: Optional[List[List[bool]]] = None,
        episode_returns: Optional[List[float]] = None,
        episode_lengths: Optional[List[int]] = None,
    ):
        self.observations: List[List[np.ndarray]] = observations or []
        self.actions: List[List[np.ndarray]] = actions or []
        self.rewards: List[List[float]] = rewards or []
        self.last_observations: List[np.ndarray] = last_observations or []
        self.dones: List[List[bool]] = dones or []
        self.episode_returns: List[float] = episode_returns or []
        self.episode_lengths: List[int] = episode_lengths or []

    # ------------------------------------------------------------------
    # Derived views (reference: experience.py:42-84)
    # ------------------------------------------------------------------
    @property
    def observations_with_last_observation(self) -> List[List[np.ndarray]]:
        return [
            list(obs) + [last]
            for obs, last in zip(self.observations, self.last_observations)
        ]

    @property
    def next_observations(self) -> List[List[np.ndarray]]:
        return [
            list(obs[1:]) + [last]
            for obs, last in zip(self.observations, self.last_observations)
        ]

    @property
    def episode_dones(self) -> List[bool]:
        return [bool(ep[-1]) for ep in self.dones]

    @property
    def flattened_observations(self) -> List[np.ndarray]:
        return [o for ep in self.observations for o in ep]

    @property
    def flattened_actions(self) -> List[np.ndarray]:
        return [a for ep in self.actions for a in ep]

    @property
    def flattened_rewards(self) -> List[float]:
        return [r for ep in self.rewards for r in ep]

    @property
    def flattened_next_observations(self) -> List[np.ndarray]:
        return [o for ep in self.next_observations for o in ep]

    @property
    def flattened_dones(self) -> List[bool]:
        return [d for ep in self.dones for d in ep]

    # ------------------------------------------------------------------
    # MI355X device-pipeline view
    # ------------------------------------------------------------------
    def set_flat_cache(self, flat: Dict[str, np.ndarray]) -> None:
        """Install a precomputed flat view (the vectorized sampler builds
        it in O(1) numpy ops instead of per-episode concatenation)."""
        self._flat_cache = flat

    def to_flat_batch(self) -> Dict[str, np.ndarray]:
        """Contiguous arrays + episode offsets for one-shot H2D upload.

        Returns dict with:
          observations [T, O*] float32, actions [T, A*], rewards [T] f32,
          episode_offsets [N+1] int32 (episode e = rows offsets[e]:offsets[e+1]),
          episode_dones [N] bool, last_observations [N, O*] float32.
        """
        cached = getattr(self, "_flat_cache", None)
        if cached is not None:
            return cached
        lengths = np.asarray(self.episode_lengths, dtype=np.int64)
        offsets = np.zeros(len(lengths) + 1, dtype=np.int32)
        np.cumsum(lengths, out=offsets[1:])
        obs = (
            np.concatenate([np.stack(ep) for ep in self.observations]).astype(np.float32)
            if self.observations
            else np.zeros((0,), dtype=np.float32)
        )
        acts = (
            np.concatenate([np.stack(ep) for ep in self.actions])
            if self.actions
            else np.zeros((0,), dtype=np.float32)
        )
        rews = np.asarray(self.flattened_rewards, dtype=np.float32)
        return {
            "observations": obs,
            "actions": acts,
            "rewards": rews,
            "episode_offsets": offsets,
            "episode_dones": np.asarray(self.episode_dones, dtype=bool),
            "last_observations": (
                np.stack(self.last_observations).astype(np.float32)
                if self.last_observations
                else np.zeros((0,), dtype=np.float32)
            ),
        }
