"""Synthetic MuJoCo-shaped benchmark environments.

There is no MuJoCo (and no network to fetch it) in this stack; the
framework's headline benchmark (BASELINE.json: PPO HalfCheetah-v4
env-steps/sec on 1-8 MI355X GPUs) runs on *synthetic-obs rollouts* of
the exact observation/action dimensionality of the named MuJoCo-v4
task.  These envs reproduce those shapes and the 1000-step horizon
(reference trains on gymnasium MuJoCo tasks: benchmarks/run_vpg.py:28,
benchmarks/run_benchmarks.py env list) with a cheap vectorized random
linear dynamical system so that env stepping measures the framework,
not a physics engine.

Reward = -|a|^2 * 0.1 + <w, s> : action-dependent and state-dependent so
the full RL pipeline (GAE, advantage normalization, policy gradient)
exercises real, learnable structure.
"""
from __future__ import annotations

import zlib

import numpy as np

from .core import Env, EnvSpec, register
from .spaces import Box

# (obs_dim, act_dim) of the gymnasium MuJoCo v4 tasks the reference
# benchmarks on (benchmarks/run_benchmarks.py).
MUJOCO_SHAPES = {
    "HalfCheetah-v4": (17, 6),
    "Walker2d-v4": (17, 6),
    "Hopper-v4": (11, 3),
    "Ant-v4": (27, 8),
    "Swimmer-v4": (8, 2),
}


class SyntheticEnv(Env):
    """Random linear dynamical system with MuJoCo-like shapes.

    s' = tanh(A s + B a + sigma * eps);  r = <w, s'> - 0.1 |a|^2.
    A is scaled to spectral-norm ~0.95 so states stay bounded.  The
    system matrices are drawn from a fixed seed per env id, so every
    rank/process sees identical dynamics (needed for DP determinism).
    """

    def __init__(self, id: str, obs_dim: int, act_dim: int, max_episode_steps: int = 1000, noise: float = 0.05):
        super().__init__()
        # float32 throughout: the dynamics exist to exercise the RL
        # pipeline, and fp32 numpy steps are ~2x fp64 at these sizes
        self.observation_space = Box(-np.inf, np.inf, shape=(obs_dim,), dtype=np.float32)
        self.action_space = Box(-1.0, 1.0, shape=(act_dim,), dtype=np.float32)
        self.spec = EnvSpec(id, max_episode_steps=max_episode_steps)
        self.noise = np.float32(noise)
        # crc32, not hash(): str hash is per-process randomized in Python 3,
        # which would give every rank/run different dynamics
        rng = np.random.default_rng(zlib.crc32(id.encode()) % (2**31))
        A = rng.standard_normal((obs_dim, obs_dim))
        # scale A to spectral norm 0.95 for bounded dynamics
        s = np.linalg.svd(A, compute_uv=False)[0]
        self.A = ((0.95 / s) * A).astype(np.float32)
        self.B = (rng.standard_normal((act_dim, obs_dim)) / np.sqrt(act_dim)).astype(np.float32)
        self.w = (rng.standard_normal(obs_dim) / np.sqrt(obs_dim)).astype(np.float32)
        self.state: np.ndarray = np.zeros((0, obs_dim), dtype=np.float32)

    def _init_state(self, n: int) -> np.ndarray:
        return (self.np_random.standard_normal((n, self.A.shape[0])) * 0.1).astype(np.float32)

    def _reset_b(self, batch: int) -> np.ndarray:
        self.state = self._init_state(batch)
        return self.state

    def _reset_idx(self, idx: np.ndarray) -> np.ndarray:
        self.state[idx] = self._init_state(len(idx))
        return self.state[idx]

    def _step_b(self, actions: np.ndarray):
        a = np.clip(np.asarray(actions, dtype=np.float32).reshape(len(self.state), -1), -1.0, 1.0)
        eps = self.np_random.standard_normal(self.state.shape).astype(np.float32)
        self.state = np.tanh(self.state @ self.A + a @ self.B + self.noise * eps)
        reward = self.state @ self.w - 0.1 * np.sum(a * a, axis=1)
        terminated = np.zeros(len(self.state), dtype=bool)
        return self.state, reward, terminated


def _register_synthetic() -> None:
    for env_id, (obs_dim, act_dim) in MUJOCO_SHAPES.items():
        register(env_id, lambda env_id=env_id, obs_dim=obs_dim, act_dim=act_dim, **kw: SyntheticEnv(env_id, obs_dim, act_dim, **kw))


_register_synthetic()
