"""MetricsManager + Evaluator unit coverage."""
import csv
import os

import numpy as np

from rl_replicas_amd import envs
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.metrics_manager import MetricsManager
from rl_replicas_amd.policies import RandomPolicy


class TestMetricsManager:
    def test_stdout_format_and_csv(self, tmp_path, capsys):
        m = MetricsManager(str(tmp_path))
        m.record_scalar("epoch", 3)
        m.record_scalar("sampling/average_episode_return", 123.456, 1000, tensorboard=True)
        m.dump()
        out = capsys.readouterr().out
        # reference stdout format: "{tag}: {:<8.3g}" (metrics_manager.py:30)
        assert "epoch: 3" in out
        assert "sampling/average_episode_return: 123" in out
        rows = list(csv.DictReader(open(tmp_path / "metrics.csv")))
        assert rows[0]["tag"] == "sampling/average_episode_return"
        assert rows[0]["step"] == "1000"
        m.close()

    def test_phase_timers(self, tmp_path, capsys):
        m = MetricsManager(str(tmp_path))
        m.record_phase_ms("sample", 1.5)
        m.record_phase_ms("sample", 2.5)
        m.record_phase_ms("train", 7.0)
        m.dump_phases(100)
        out = capsys.readouterr().out
        assert "phase_ms/sample: 4" in out
        assert "phase_ms/train: 7" in out
        m.close()


class TestEvaluator:
    def test_vectorized_matches_protocol(self):
        env = envs.make("Pendulum-v1")
        env.action_space.seed(0)
        returns, lengths = Evaluator(seed=0).evaluate(RandomPolicy(env.action_space), env, 4)
        assert len(returns) == len(lengths) == 4
        assert all(l == 200 for l in lengths)  # Pendulum truncates at 200
        assert all(r < 0 for r in returns)

    def test_serial_fallback_for_non_batched_envs(self):
        class SerialOnly:
            """Third-party-style env without the batched protocol."""

            def __init__(self):
                self.inner = envs.make("Pendulum-v1")
                self.action_space = self.inner.action_space
                self.observation_space = self.inner.observation_space
                self.spec = self.inner.spec

            def reset(self, **kw):
                return self.inner.reset(**kw)

            def step(self, a):
                return self.inner.step(a)

        env = SerialOnly()
        env.action_space.seed(0)
        returns, lengths = Evaluator(seed=0).evaluate(RandomPolicy(env.action_space), env, 2)
        assert len(returns) == 2 and all(l == 200 for l in lengths)

    def test_deterministic_given_seed(self):
        def run():
            env = envs.make("Pendulum-v1")
            env.action_space.seed(7)
            return Evaluator(seed=7).evaluate(RandomPolicy(env.action_space), env, 3)[0]

        assert run() == run()
