"""Policy evaluator.

API parity: reference src/rl_replicas/evaluator.py:9-52 — runs
`num_episodes` full episodes in a separate env and returns
`(episode_returns, episode_lengths)`; seeds the env on the first reset
only.  Stochastic policies SAMPLE during evaluation (the reference has
no deterministic flag); DDPG/TD3 pass the un-noised deterministic
policy (reference ddpg.py:155).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

from rl_replicas_amd.policies import Policy


class Evaluator:
    def __init__(self, seed: Optional[int] = None, vectorized: bool = True):
        self.seed = seed
        self.vectorized = vectorized

    def evaluate(self, policy: Policy, env, num_episodes: int) -> Tuple[List[float], List[int]]:
        if self.vectorized and hasattr(env, "_reset_b"):
            return self._evaluate_vectorized(policy, env, num_episodes)
        return self._evaluate_serial(policy, env, num_episodes)

    # serial reference path (third-party envs)
    def _evaluate_serial(self, policy: Policy, env, num_episodes: int):
        episode_returns: List[float] = []
        episode_lengths: List[int] = []

        observation, _ = env.reset(seed=self.seed)
        for _ in range(num_episodes):
            done = False
            ep_return = 0.0
            ep_length = 0
            while not done:
                action = policy.get_action_numpy(observation)
                observation, reward, terminated, truncated, _ = env.step(action)
                done = bool(terminated or truncated)
                ep_return += float(reward)
                ep_length += 1
            observation, _ = env.reset()
            episode_returns.append(ep_return)
            episode_lengths.append(ep_length)

        return episode_returns, episode_lengths

    # MI355X fast path: all episodes advance together — one batched
    # policy forward per step instead of num_episodes serial rollouts
    # (episodes are independent, so the returns distribution is the
    # reference's; only the RNG consumption pattern differs)
    def _evaluate_vectorized(self, policy: Policy, env, num_episodes: int):
        import numpy as np

        from rl_replicas_amd.envs.vector import VectorEnv

        venv = VectorEnv(env, num_episodes)
        # reseed on EVERY call, like the reference's reset(seed=self.seed)
        # (evaluator.py:30) and our own serial path: periodic evaluations
        # start from the same seeded initial states each time
        obs = venv.reset(seed=self.seed)
        returns = np.zeros(num_episodes)
        lengths = np.zeros(num_episodes, dtype=np.int64)
        finished = np.zeros(num_episodes, dtype=bool)
        max_steps = env.spec.max_episode_steps or 100_000
        for _ in range(max_steps):
            actions = np.asarray(policy.get_action_numpy(obs))
            obs, rewards, terminated, truncated, _ = venv.step(actions)
            active = ~finished
            returns[active] += rewards[active]
            lengths[active] += 1
            finished |= terminated | truncated
            if finished.all():
                break
        return returns.tolist(), lengths.tolist()
