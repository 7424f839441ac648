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
    def __init__(self, seed: Optional[int] = None):
        self.seed = seed

    def evaluate(self, policy: Policy, env, num_episodes: int) -> Tuple[List[float], List[int]]:
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
