"""Vectorized environment driver.

The reference steps ONE env serially (batch_sampler.py:55-99) — the
whole-node throughput ceiling of that design is one CPU core.  The
MI355X-first design batches N env instances behind a single object so
each sampler step produces an obs batch [N, obs_dim] that crosses to
the GPU once, and the policy MLP runs one batched forward per step.

Native batched envs (CartPole/Pendulum/Synthetic implement `_step_b`
over a state matrix) run fully vectorized in numpy — no per-instance
Python loop, no worker processes needed at these state sizes.

Autoreset semantics: `step()` returns the *post-autoreset* observation
in `obs`, and the true successor observation (pre-reset for done
instances) in `final_obs`, which is what GAE bootstrapping needs.
"""
from __future__ import annotations

from typing import Callable, Optional, Tuple, Union

import numpy as np

from .core import Env, make


class VectorEnv:
    def __init__(self, env: Union[str, Callable[[], Env], Env], num_envs: int, **make_kwargs):
        if isinstance(env, str):
            self.env = make(env, **make_kwargs)
        elif callable(env) and not isinstance(env, Env):
            self.env = env()
        else:
            self.env = env
        if not hasattr(self.env, "_step_b"):
            raise TypeError(
                f"VectorEnv requires a natively batched env (got {type(self.env).__name__}); "
                "wrap serial third-party envs with SerialVectorEnv instead"
            )
        self.num_envs = int(num_envs)
        self.observation_space = self.env.observation_space
        self.action_space = self.env.action_space
        self.spec = self.env.spec
        self._elapsed = np.zeros(self.num_envs, dtype=np.int64)

    def seed(self, seed: Optional[int]) -> None:
        # same seeding contract as Env.reset(seed=...): env dynamics RNG
        # plus a derived action-space stream
        self.env._np_random = np.random.default_rng(seed)
        if seed is not None:
            self.action_space.seed(seed + 1000)

    def reset(self, *, seed: Optional[int] = None) -> np.ndarray:
        if seed is not None:
            self.seed(seed)
        self._elapsed[:] = 0
        return self.env._reset_b(self.num_envs)

    def step(self, actions: np.ndarray) -> Tuple[np.ndarray, np.ndarray, np.ndarray, np.ndarray, np.ndarray]:
        obs, reward, terminated = self.env._step_b(actions)
        self._elapsed += 1
        max_steps = self.spec.max_episode_steps
        truncated = (
            (self._elapsed >= max_steps) & ~terminated
            if max_steps is not None
            else np.zeros(self.num_envs, dtype=bool)
        )
        final_obs = obs
        done = terminated | truncated
        if done.any():
            idx = np.nonzero(done)[0]
            final_obs = obs.copy()
            new_obs = self.env._reset_idx(idx)
            obs[idx] = new_obs
            self._elapsed[idx] = 0
        return obs, reward, terminated, truncated, final_obs

    def close(self) -> None:
        self.env.close()


class SerialVectorEnv:
    """Fallback vectorization over N independent serial envs.

    Used for third-party (e.g. real gymnasium) envs that only expose
    the scalar step API.  Same interface as VectorEnv.
    """

    def __init__(self, env_fns):
        self.envs = [fn() for fn in env_fns]
        self.num_envs = len(self.envs)
        e = self.envs[0]
        self.observation_space = e.observation_space
        self.action_space = e.action_space
        self.spec = e.spec

    def seed(self, seed: Optional[int]) -> None:
        self._seeds = [None if seed is None else seed + i for i in range(self.num_envs)]

    def reset(self, *, seed: Optional[int] = None):
        if seed is not None:
            self.seed(seed)
        seeds = getattr(self, "_seeds", [None] * self.num_envs)
        obs = [e.reset(seed=s)[0] for e, s in zip(self.envs, seeds)]
        self._seeds = [None] * self.num_envs
        return np.stack(obs).astype(np.float32)

    def step(self, actions: np.ndarray):
        obs_l, rew_l, term_l, trunc_l, final_l = [], [], [], [], []
        for e, a in zip(self.envs, actions):
            o, r, te, tr, _ = e.step(a)
            final_l.append(o)
            if te or tr:
                o, _ = e.reset()
            obs_l.append(o)
            rew_l.append(r)
            term_l.append(te)
            trunc_l.append(tr)
        return (
            np.stack(obs_l).astype(np.float32),
            np.asarray(rew_l, dtype=np.float64),
            np.asarray(term_l, dtype=bool),
            np.asarray(trunc_l, dtype=bool),
            np.stack(final_l).astype(np.float32),
        )

    def close(self) -> None:
        for e in self.envs:
            e.close()
