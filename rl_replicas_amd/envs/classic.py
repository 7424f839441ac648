"""Classic-control environments (native, batched implementations).

CartPole-v1 and Pendulum-v1 with the exact published dynamics the
reference trains on through gymnasium (its CI envs — reference:
tests/test_ppo.py:36-71, tests/test_td3.py uses Pendulum-v1).  The
physics follow the standard gymnasium equations; state transitions are
vectorized over a leading batch dimension so one object can drive
thousands of instances for the GPU-feeding vector sampler.
"""
from __future__ import annotations

import numpy as np

from .core import Env, EnvSpec, register
from .spaces import Box, Discrete


class CartPoleEnv(Env):
    """CartPole-v1: discrete(2) actions, 4-dim obs, reward 1/step.

    Dynamics: Barto-Sutton-Anderson cartpole with Euler integration,
    identical constants to gymnasium's CartPole-v1 (termination at
    |x|>2.4 or |theta|>12deg, 500-step time limit).
    """

    GRAVITY = 9.8
    MASSCART = 1.0
    MASSPOLE = 0.1
    TOTAL_MASS = MASSCART + MASSPOLE
    LENGTH = 0.5  # half the pole's length
    POLEMASS_LENGTH = MASSPOLE * LENGTH
    FORCE_MAG = 10.0
    TAU = 0.02
    THETA_THRESHOLD = 12 * 2 * np.pi / 360
    X_THRESHOLD = 2.4

    def __init__(self) -> None:
        super().__init__()
        high = np.array(
            [self.X_THRESHOLD * 2, np.finfo(np.float32).max, self.THETA_THRESHOLD * 2, np.finfo(np.float32).max],
            dtype=np.float32,
        )
        self.observation_space = Box(-high, high, dtype=np.float32)
        self.action_space = Discrete(2)
        self.spec = EnvSpec("CartPole-v1", max_episode_steps=500)
        self.state: np.ndarray = np.zeros((0, 4), dtype=np.float64)

    def _init_state(self, n: int) -> np.ndarray:
        return self.np_random.uniform(-0.05, 0.05, size=(n, 4))

    def _reset_b(self, batch: int) -> np.ndarray:
        self.state = self._init_state(batch)
        return self.state.astype(np.float32)

    def _reset_idx(self, idx: np.ndarray) -> np.ndarray:
        self.state[idx] = self._init_state(len(idx))
        return self.state[idx].astype(np.float32)

    def _step_b(self, actions: np.ndarray):
        x, x_dot, theta, theta_dot = self.state.T
        force = np.where(np.asarray(actions).reshape(-1) == 1, self.FORCE_MAG, -self.FORCE_MAG)
        costheta = np.cos(theta)
        sintheta = np.sin(theta)
        temp = (force + self.POLEMASS_LENGTH * theta_dot**2 * sintheta) / self.TOTAL_MASS
        thetaacc = (self.GRAVITY * sintheta - costheta * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASSPOLE * costheta**2 / self.TOTAL_MASS)
        )
        xacc = temp - self.POLEMASS_LENGTH * thetaacc * costheta / self.TOTAL_MASS
        x = x + self.TAU * x_dot
        x_dot = x_dot + self.TAU * xacc
        theta = theta + self.TAU * theta_dot
        theta_dot = theta_dot + self.TAU * thetaacc
        self.state = np.stack([x, x_dot, theta, theta_dot], axis=1)
        terminated = (np.abs(x) > self.X_THRESHOLD) | (np.abs(theta) > self.THETA_THRESHOLD)
        reward = np.ones_like(x, dtype=np.float64)
        return self.state.astype(np.float32), reward, terminated


def _angle_normalize(x: np.ndarray) -> np.ndarray:
    return ((x + np.pi) % (2 * np.pi)) - np.pi


class PendulumEnv(Env):
    """Pendulum-v1: 1-dim torque action in [-2, 2], 3-dim obs, 200-step limit."""

    MAX_SPEED = 8.0
    MAX_TORQUE = 2.0
    DT = 0.05
    G = 10.0
    M = 1.0
    L = 1.0

    def __init__(self) -> None:
        super().__init__()
        high = np.array([1.0, 1.0, self.MAX_SPEED], dtype=np.float32)
        self.observation_space = Box(-high, high, dtype=np.float32)
        self.action_space = Box(-self.MAX_TORQUE, self.MAX_TORQUE, shape=(1,), dtype=np.float32)
        self.spec = EnvSpec("Pendulum-v1", max_episode_steps=200)
        self.state: np.ndarray = np.zeros((0, 2), dtype=np.float64)  # [theta, theta_dot]

    def _obs(self) -> np.ndarray:
        th, thdot = self.state.T
        return np.stack([np.cos(th), np.sin(th), thdot], axis=1).astype(np.float32)

    def _init_state(self, n: int) -> np.ndarray:
        high = np.array([np.pi, 1.0])
        return self.np_random.uniform(-high, high, size=(n, 2))

    def _reset_b(self, batch: int) -> np.ndarray:
        self.state = self._init_state(batch)
        return self._obs()

    def _reset_idx(self, idx: np.ndarray) -> np.ndarray:
        self.state[idx] = self._init_state(len(idx))
        return self._obs()[idx]

    def _step_b(self, actions: np.ndarray):
        th, thdot = self.state.T
        u = np.clip(np.asarray(actions, dtype=np.float64).reshape(len(th), -1)[:, 0], -self.MAX_TORQUE, self.MAX_TORQUE)
        costs = _angle_normalize(th) ** 2 + 0.1 * thdot**2 + 0.001 * u**2
        newthdot = thdot + (3 * self.G / (2 * self.L) * np.sin(th) + 3.0 / (self.M * self.L**2) * u) * self.DT
        newthdot = np.clip(newthdot, -self.MAX_SPEED, self.MAX_SPEED)
        newth = th + newthdot * self.DT
        self.state = np.stack([newth, newthdot], axis=1)
        terminated = np.zeros(len(th), dtype=bool)
        return self._obs(), -costs, terminated


register("CartPole-v1", CartPoleEnv)
register("Pendulum-v1", PendulumEnv)
