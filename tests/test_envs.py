"""Native env layer tests: spaces, dynamics, vectorization, registry."""
import numpy as np
import pytest

from rl_replicas_amd import envs


class TestSpaces:
    def test_box_sample_bounds(self):
        box = envs.Box(-2.0, 2.0, shape=(3,), seed=0)
        for _ in range(100):
            s = box.sample()
            assert s.shape == (3,)
            assert np.all(s >= -2.0) and np.all(s <= 2.0)

    def test_discrete_sample(self):
        d = envs.Discrete(4, seed=0)
        samples = {int(d.sample()) for _ in range(200)}
        assert samples == {0, 1, 2, 3}

    def test_contains(self):
        box = envs.Box(-1.0, 1.0, shape=(2,))
        assert box.contains(np.zeros(2, dtype=np.float32))
        assert not box.contains(np.full(2, 5.0, dtype=np.float32))
        d = envs.Discrete(3)
        assert d.contains(2) and not d.contains(3)


class TestRegistry:
    def test_make_known(self):
        env = envs.make("CartPole-v1")
        assert env.spec.id == "CartPole-v1"
        assert env.spec.max_episode_steps == 500

    def test_make_unknown(self):
        with pytest.raises(KeyError):
            envs.make("DoesNotExist-v0")

    def test_mujoco_shapes(self):
        for env_id, (obs_dim, act_dim) in envs.MUJOCO_SHAPES.items():
            env = envs.make(env_id)
            assert env.observation_space.shape == (obs_dim,)
            assert env.action_space.shape == (act_dim,)


class TestCartPole:
    def test_episode_runs_and_terminates(self):
        env = envs.make("CartPole-v1")
        obs, _ = env.reset(seed=0)
        assert obs.shape == (4,)
        steps = 0
        terminated = truncated = False
        while not (terminated or truncated):
            obs, reward, terminated, truncated, _ = env.step(env.action_space.sample())
            assert reward == 1.0
            steps += 1
            assert steps <= 500
        # random policy should fall over well before the time limit
        assert terminated and steps < 500

    def test_determinism(self):
        def rollout():
            env = envs.make("CartPole-v1")
            env.action_space.seed(123)
            obs, _ = env.reset(seed=42)
            traj = [obs]
            for _ in range(20):
                obs, *_ = env.step(env.action_space.sample())
                traj.append(obs)
            return np.stack(traj)

        np.testing.assert_array_equal(rollout(), rollout())


class TestPendulum:
    def test_truncates_at_200(self):
        env = envs.make("Pendulum-v1")
        obs, _ = env.reset(seed=0)
        assert obs.shape == (3,)
        for t in range(200):
            obs, reward, terminated, truncated, _ = env.step(np.array([0.5], dtype=np.float32))
            assert reward <= 0.0
            assert not terminated
        assert truncated

    def test_obs_is_cos_sin_thdot(self):
        env = envs.make("Pendulum-v1")
        obs, _ = env.reset(seed=0)
        assert abs(obs[0] ** 2 + obs[1] ** 2 - 1.0) < 1e-5


class TestSynthetic:
    def test_bounded_and_identical_dynamics_across_instances(self):
        e1 = envs.make("HalfCheetah-v4")
        e2 = envs.make("HalfCheetah-v4")
        np.testing.assert_array_equal(e1.A, e2.A)  # seeded by env id
        obs, _ = e1.reset(seed=0)
        for _ in range(50):
            obs, r, term, trunc, _ = e1.step(e1.action_space.sample())
            assert np.all(np.abs(obs) <= 1.0)
            assert not term


class TestVectorEnv:
    def test_batch_shapes_and_autoreset(self):
        venv = envs.VectorEnv("CartPole-v1", num_envs=8)
        obs = venv.reset(seed=0)
        assert obs.shape == (8, 4)
        saw_done = False
        for _ in range(300):
            actions = np.random.randint(0, 2, size=8)
            obs, rew, term, trunc, final_obs = venv.step(actions)
            assert obs.shape == (8, 4) and rew.shape == (8,)
            if term.any():
                saw_done = True
                # autoreset: post-reset obs is near the origin while the
                # final_obs of a terminated cartpole is out of bounds
                i = int(np.nonzero(term)[0][0])
                assert np.all(np.abs(obs[i]) <= 0.05 + 1e-6)
                assert (
                    abs(final_obs[i][0]) > 2.4 or abs(final_obs[i][2]) > 12 * 2 * np.pi / 360
                )
        assert saw_done

    def test_vector_matches_serial_cartpole(self):
        """B=1 vectorized stepping reproduces the serial env exactly."""
        serial = envs.make("CartPole-v1")
        obs_s, _ = serial.reset(seed=7)
        venv = envs.VectorEnv("CartPole-v1", num_envs=1)
        obs_v = venv.reset(seed=7)
        np.testing.assert_allclose(obs_s, obs_v[0])
        for _ in range(30):
            a = 1
            obs_s, r_s, te_s, tr_s, _ = serial.step(a)
            obs_v, r_v, te_v, tr_v, _ = venv.step(np.array([a]))
            if te_s or tr_s:
                break
            np.testing.assert_allclose(obs_s, obs_v[0], rtol=1e-6)
            assert r_s == r_v[0]
