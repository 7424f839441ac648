"""Native environment API + registry.

The reference consumes `gymnasium.Env` objects (`env.step/reset`,
`env.action_space`, `env.observation_space`, `env.spec.id` — see
src/rl_replicas/samplers/batch_sampler.py:49-64, algorithms/ddpg.py:66).
gymnasium is not installed in this stack, so this module defines the
same 5-tuple step API natively, plus a `make()` registry so user code
written against the reference (`gym.make("CartPole-v1")` →
`rl_replicas_amd.envs.make("CartPole-v1")`) ports over directly.

Design difference from the reference's world (MI355X-first): every env
here also implements a *batched* interface (`reset_batch/step_batch`
over B instances at once) used by the vectorized sampler that feeds the
GPU learner — the serial `step()` API is a B=1 view over it.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Optional, Tuple

import numpy as np

from .spaces import Space


class EnvSpec:
    def __init__(self, id: str, max_episode_steps: Optional[int] = None):
        self.id = id
        self.max_episode_steps = max_episode_steps

    def __repr__(self) -> str:
        return f"EnvSpec({self.id!r}, max_episode_steps={self.max_episode_steps})"


class Env:
    """Serial environment: gymnasium-compatible 5-tuple step API.

    Subclasses implement the *batched* `_reset_b` / `_step_b` methods
    (vectorized over a leading batch dim); this base class exposes the
    scalar gymnasium API as batch-size-1 over them, and applies the
    time limit (`spec.max_episode_steps`) -> `truncated`.
    """

    observation_space: Space
    action_space: Space
    spec: EnvSpec

    def __init__(self) -> None:
        self._np_random: Optional[np.random.Generator] = None
        self._elapsed_steps = 0

    # -- RNG ---------------------------------------------------------------
    @property
    def np_random(self) -> np.random.Generator:
        if self._np_random is None:
            self._np_random = np.random.default_rng()
        return self._np_random

    # -- batched core (implemented by subclasses) --------------------------
    def _reset_b(self, batch: int) -> np.ndarray:
        """Reset `batch` instances; returns obs[batch, obs_dim]."""
        raise NotImplementedError

    def _step_b(self, actions: np.ndarray) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
        """Step all instances; returns (obs[B,D], reward[B], terminated[B])."""
        raise NotImplementedError

    # -- serial gymnasium API ---------------------------------------------
    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        if seed is not None:
            self._np_random = np.random.default_rng(seed)
            # derive the action-space stream from the same seed so that
            # random-policy warm-up (DDPG/TD3) is deterministic
            self.action_space.seed(seed + 1000)
        self._elapsed_steps = 0
        obs = self._reset_b(1)[0]
        return obs, {}

    def step(self, action):
        a = np.asarray(action)
        obs, reward, terminated = self._step_b(a[None] if a.shape == self.action_space.shape else a.reshape((1,) + self.action_space.shape))
        self._elapsed_steps += 1
        truncated = bool(
            self.spec.max_episode_steps is not None and self._elapsed_steps >= self.spec.max_episode_steps
        )
        return obs[0], float(reward[0]), bool(terminated[0]), truncated, {}

    def render(self):  # pragma: no cover - no rendering backend
        return None

    def close(self) -> None:
        return None


# ---------------------------------------------------------------------------
# Registry
# ---------------------------------------------------------------------------
_REGISTRY: Dict[str, Callable[..., Env]] = {}


def register(id: str, entry_point: Callable[..., Env]) -> None:
    _REGISTRY[id] = entry_point


def make(id: str, **kwargs: Any) -> Env:
    """Create a registered environment by id (gym.make equivalent)."""
    if id not in _REGISTRY:
        raise KeyError(
            f"Unknown environment id {id!r}. Registered: {sorted(_REGISTRY)}"
        )
    return _REGISTRY[id](**kwargs)


def registered_ids():
    return sorted(_REGISTRY)
