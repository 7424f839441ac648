"""Action/observation space primitives.

gymnasium is not a dependency of this framework: the reference
(rl_replicas) drives `gymnasium.Env`/`gymnasium.spaces` objects
(reference: src/rl_replicas/samplers/batch_sampler.py:24,
policies/random_policy.py:19-27), so we provide an API-compatible
native space layer (`Box`, `Discrete`) that the samplers, policies and
algorithms consume.  Any object with the same duck-typed surface
(`shape`, `dtype`, `sample()`, `seed()`, and `n`/`low`/`high`) works,
including real gymnasium spaces if the user has gymnasium installed.
"""
from __future__ import annotations

from typing import Optional, Sequence, Tuple, Union

import numpy as np


class Space:
    """Base class of observation/action spaces.

    Mirrors the `gymnasium.Space` surface used by the reference:
    `sample()` (random_policy.py:27) and `shape`/`dtype`.
    """

    def __init__(self, shape: Optional[Tuple[int, ...]] = None, dtype=None, seed: Optional[int] = None):
        self.shape = tuple(shape) if shape is not None else None
        self.dtype = np.dtype(dtype) if dtype is not None else None
        self._np_random: Optional[np.random.Generator] = None
        if seed is not None:
            self.seed(seed)

    @property
    def np_random(self) -> np.random.Generator:
        if self._np_random is None:
            self._np_random = np.random.default_rng()
        return self._np_random

    def seed(self, seed: Optional[int] = None) -> None:
        self._np_random = np.random.default_rng(seed)

    def sample(self):
        raise NotImplementedError

    def contains(self, x) -> bool:
        raise NotImplementedError


class Box(Space):
    """Continuous box space: elementwise bounds [low, high]."""

    def __init__(
        self,
        low: Union[float, np.ndarray],
        high: Union[float, np.ndarray],
        shape: Optional[Sequence[int]] = None,
        dtype=np.float32,
        seed: Optional[int] = None,
    ):
        if shape is None:
            shape = np.broadcast(np.asarray(low), np.asarray(high)).shape
        shape = tuple(int(s) for s in shape)
        super().__init__(shape, dtype, seed)
        self.low = np.broadcast_to(np.asarray(low, dtype=self.dtype), shape).copy()
        self.high = np.broadcast_to(np.asarray(high, dtype=self.dtype), shape).copy()
        self.bounded_below = np.isfinite(self.low)
        self.bounded_above = np.isfinite(self.high)

    def sample(self, batch: Optional[int] = None) -> np.ndarray:
        # Bounded dims: uniform in [low, high); unbounded: standard normal
        # (matches gymnasium.Box.sample semantics closely enough for
        # exploration warm-up; the reference only uses bounded Boxes).
        # `batch` (extension): draw a vectorized [batch, *shape] sample in
        # one call — used by RandomPolicy over vectorized envs.
        shape = self.shape if batch is None else (batch,) + self.shape
        out = np.empty(shape, dtype=np.float64)
        bounded = np.broadcast_to(self.bounded_below & self.bounded_above, shape)
        u = self.np_random.random(shape)
        out[bounded] = (self.low + u * (self.high - self.low))[bounded]
        if not bounded.all():
            n = self.np_random.standard_normal(shape)
            only_below = np.broadcast_to(self.bounded_below & ~self.bounded_above, shape)
            only_above = np.broadcast_to(~self.bounded_below & self.bounded_above, shape)
            neither = np.broadcast_to(~self.bounded_below & ~self.bounded_above, shape)
            out[only_below] = (self.low + np.abs(n))[only_below]
            out[only_above] = (self.high - np.abs(n))[only_above]
            out[neither] = n[neither]
        return out.astype(self.dtype)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return bool(x.shape == self.shape and np.all(x >= self.low - 1e-6) and np.all(x <= self.high + 1e-6))

    def __repr__(self) -> str:
        return f"Box({self.low.min()}, {self.high.max()}, {self.shape}, {self.dtype})"


class Discrete(Space):
    """Finite set {0, ..., n-1}."""

    def __init__(self, n: int, seed: Optional[int] = None, start: int = 0):
        super().__init__((), np.int64, seed)
        self.n = int(n)
        self.start = int(start)

    def sample(self, batch: Optional[int] = None):
        if batch is not None:
            return self.start + self.np_random.integers(self.n, size=batch)
        return np.int64(self.start + self.np_random.integers(self.n))

    def contains(self, x) -> bool:
        try:
            xi = int(x)
        except (TypeError, ValueError):
            return False
        return self.start <= xi < self.start + self.n

    def __repr__(self) -> str:
        return f"Discrete({self.n})"
