"""Golden-replay sampler tests (strategy from reference
tests/test_samplers.py:26-109): re-run the identical seeded env +
random-policy loop by hand and assert the sampler's flattened arrays
match exactly; plus the continuous-sampling equivalence test and the
vectorized-sampler consistency checks."""
import numpy as np
import pytest

from rl_replicas_amd import envs
from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import RandomPolicy
from rl_replicas_amd.samplers import BatchSampler, VectorSampler

NUM_SAMPLES = 200
ENV_SEED = 5


def manual_rollout(num_samples: int):
    """Hand-rolled oracle of the BatchSampler contract."""
    env = envs.make("CartPole-v1")
    env.action_space.seed(ENV_SEED)
    obs, _ = env.reset(seed=ENV_SEED)
    observations, actions, rewards, dones = [], [], [], []
    last_observations, episode_returns, episode_lengths = [], [], []
    ep_obs, ep_act, ep_rew, ep_done = [], [], [], []
    for step in range(num_samples):
        a = env.action_space.sample()
        ep_obs.append(obs)
        ep_act.append(np.asarray(a))
        obs, r, term, trunc, _ = env.step(a)
        done = term or trunc
        ep_rew.append(r)
        ep_done.append(done)
        if done or step == num_samples - 1:
            observations.append(ep_obs)
            actions.append(ep_act)
            rewards.append(ep_rew)
            dones.append(ep_done)
            last_observations.append(obs)
            episode_returns.append(sum(ep_rew))
            episode_lengths.append(len(ep_rew))
            if done:
                obs, _ = env.reset()
            ep_obs, ep_act, ep_rew, ep_done = [], [], [], []
    return Experience(
        observations, actions, rewards, last_observations, dones, episode_returns, episode_lengths
    )


def test_batch_sampler_matches_golden_replay():
    env = envs.make("CartPole-v1")
    env.action_space.seed(ENV_SEED)
    policy = RandomPolicy(env.action_space)
    sampler = BatchSampler(env, seed=ENV_SEED)
    experience = sampler.sample(NUM_SAMPLES, policy)

    oracle = manual_rollout(NUM_SAMPLES)

    np.testing.assert_array_equal(
        np.stack(experience.flattened_observations), np.stack(oracle.flattened_observations)
    )
    np.testing.assert_array_equal(
        np.stack(experience.flattened_actions), np.stack(oracle.flattened_actions)
    )
    np.testing.assert_array_equal(experience.flattened_rewards, oracle.flattened_rewards)
    np.testing.assert_array_equal(experience.flattened_dones, oracle.flattened_dones)
    np.testing.assert_array_equal(np.stack(experience.last_observations), np.stack(oracle.last_observations))
    assert experience.episode_returns == oracle.episode_returns
    assert experience.episode_lengths == oracle.episode_lengths


def test_continuous_sampling_equals_one_long_rollout():
    """10x sample(100) with is_continuous=True == one sample(1000)
    (reference test_samplers.py:85-109)."""

    def collect(chunks):
        env = envs.make("CartPole-v1")
        env.action_space.seed(ENV_SEED)
        policy = RandomPolicy(env.action_space)
        sampler = BatchSampler(env, seed=ENV_SEED, is_continuous=True)
        parts = [sampler.sample(n, policy) for n in chunks]
        obs = [o for p in parts for o in p.flattened_observations]
        acts = [a for p in parts for a in p.flattened_actions]
        rews = [r for p in parts for r in p.flattened_rewards]
        return np.stack(obs), np.stack(acts), np.asarray(rews)

    o1, a1, r1 = collect([100] * 10)
    o2, a2, r2 = collect([1000])
    np.testing.assert_array_equal(o1, o2)
    np.testing.assert_array_equal(a1, a2)
    np.testing.assert_array_equal(r1, r2)


class TestVectorSampler:
    def test_totals_and_structure(self):
        venv = envs.VectorEnv("CartPole-v1", num_envs=4)
        policy = RandomPolicy(venv.action_space)
        venv.action_space.seed(0)
        sampler = VectorSampler(venv, seed=0)
        exp = sampler.sample(400, policy)
        assert sum(exp.episode_lengths) == 400
        assert len(exp.observations) == len(exp.last_observations) == len(exp.episode_returns)
        # ragged structure is consistent
        for ep_obs, ep_act, ep_rew, ep_done, L in zip(
            exp.observations, exp.actions, exp.rewards, exp.dones, exp.episode_lengths
        ):
            assert len(ep_obs) == len(ep_act) == len(ep_rew) == len(ep_done) == L
        # flat batch offsets line up
        flat = exp.to_flat_batch()
        assert flat["observations"].shape == (400, 4)
        assert flat["episode_offsets"][-1] == 400

    def test_single_env_vector_sampler_matches_batch_sampler(self):
        """VectorSampler over 1 instance == BatchSampler on the serial env."""
        venv = envs.VectorEnv("CartPole-v1", num_envs=1)
        venv.action_space.seed(ENV_SEED)
        vs = VectorSampler(venv, seed=ENV_SEED)
        exp_v = vs.sample(NUM_SAMPLES, RandomPolicy(venv.action_space))

        env = envs.make("CartPole-v1")
        env.action_space.seed(ENV_SEED)
        bs = BatchSampler(env, seed=ENV_SEED)
        exp_b = bs.sample(NUM_SAMPLES, RandomPolicy(env.action_space))

        np.testing.assert_allclose(
            np.stack(exp_v.flattened_observations),
            np.stack(exp_b.flattened_observations),
            rtol=1e-6,
        )
        assert exp_v.episode_lengths == exp_b.episode_lengths
        assert exp_v.episode_dones == exp_b.episode_dones

    def test_rejects_indivisible_batch(self):
        venv = envs.VectorEnv("CartPole-v1", num_envs=3)
        sampler = VectorSampler(venv, seed=0)
        with pytest.raises(ValueError):
            sampler.sample(100, RandomPolicy(venv.action_space))


def test_vector_sampler_flat_cache_matches_recompute():
    """The O(1) flat view the sampler caches equals the per-episode
    concatenation path."""
    venv = envs.VectorEnv("CartPole-v1", num_envs=5)
    venv.action_space.seed(3)
    s = VectorSampler(venv, seed=3)
    exp = s.sample(200, RandomPolicy(venv.action_space))
    cached = exp.to_flat_batch()
    exp._flat_cache = None
    recomputed = exp.to_flat_batch()
    for key in recomputed:
        np.testing.assert_array_equal(
            np.asarray(cached[key], dtype=np.asarray(recomputed[key]).dtype),
            recomputed[key],
        )
