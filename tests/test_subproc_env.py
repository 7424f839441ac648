"""SubprocVectorEnv: worker-process vectorization over serial envs."""
import numpy as np
import pytest

from rl_replicas_amd import envs
from rl_replicas_amd.policies import RandomPolicy
from rl_replicas_amd.samplers import VectorSampler


@pytest.fixture
def subproc_cartpole():
    venv = envs.SubprocVectorEnv(
        [lambda: envs.make("CartPole-v1") for _ in range(6)], num_workers=3
    )
    yield venv
    venv.close()


def test_step_shapes_and_autoreset(subproc_cartpole):
    venv = subproc_cartpole
    obs = venv.reset(seed=0)
    assert obs.shape == (6, 4)
    saw_done = False
    for _ in range(300):
        actions = np.random.randint(0, 2, size=6)
        obs, rew, term, trunc, final = venv.step(actions)
        assert obs.shape == (6, 4) and rew.shape == (6,)
        if term.any():
            saw_done = True
            i = int(np.nonzero(term)[0][0])
            # autoreset happened; final holds the true terminal state
            assert np.all(np.abs(obs[i]) <= 0.0501)
            assert abs(final[i][0]) > 2.4 or abs(final[i][2]) > 12 * 2 * np.pi / 360
    assert saw_done


def test_deterministic_given_seed():
    def rollout():
        venv = envs.SubprocVectorEnv(
            [lambda: envs.make("CartPole-v1") for _ in range(4)], num_workers=2
        )
        try:
            obs = venv.reset(seed=42)
            traj = [obs]
            for t in range(20):
                obs, *_ = venv.step(np.full(4, t % 2))
                traj.append(obs)
            return np.stack(traj)
        finally:
            venv.close()

    np.testing.assert_array_equal(rollout(), rollout())


def test_drives_vector_sampler(subproc_cartpole):
    venv = subproc_cartpole
    venv.action_space.seed(0)
    sampler = VectorSampler(venv, seed=0)
    exp = sampler.sample(120, RandomPolicy(venv.action_space))
    assert sum(exp.episode_lengths) == 120
    flat = exp.to_flat_batch()
    assert flat["observations"].shape == (120, 4)
