"""Replay buffer property tests (the reference ships none — SURVEY.md §4)."""
import numpy as np
import pytest

from rl_replicas_amd.experience import Experience
from rl_replicas_amd.replay_buffer import ReplayBuffer


def make_experience(n_steps: int, obs_dim: int = 3, start: int = 0) -> Experience:
    """One episode of n_steps with recognizable values: obs[t] = start+t."""
    obs = [np.full(obs_dim, start + t, dtype=np.float32) for t in range(n_steps)]
    acts = [np.array([float(start + t)], dtype=np.float32) for t in range(n_steps)]
    rews = [float(start + t) for t in range(n_steps)]
    dones = [False] * (n_steps - 1) + [True]
    return Experience(
        [obs], [acts], [rews], [np.full(obs_dim, start + n_steps, dtype=np.float32)], [dones],
        [sum(rews)], [n_steps],
    )


def test_add_and_size():
    rb = ReplayBuffer(buffer_size=100)
    rb.add_experience(make_experience(30))
    assert len(rb) == 30
    rb.add_experience(make_experience(30, start=100))
    assert len(rb) == 60


def test_ring_overflow_keeps_newest():
    rb = ReplayBuffer(buffer_size=50)
    rb.add_experience(make_experience(40))          # values 0..39
    rb.add_experience(make_experience(40, start=100))  # values 100..139, overflows
    assert len(rb) == 50
    stored = rb._storage["rewards"][:50].numpy()
    # every one of the newest 40 transitions must still be present
    for v in range(100, 140):
        assert float(v) in stored
    # at most 10 of the oldest remain
    assert (stored < 100).sum() == 10


def test_minibatch_shapes_and_membership():
    rb = ReplayBuffer(buffer_size=100)
    rb.add_experience(make_experience(20))
    mb = rb.sample_minibatch(16)
    assert mb["observations"].shape == (16, 3)
    assert mb["actions"].shape == (16, 1)
    assert mb["rewards"].shape == (16,)
    assert mb["next_observations"].shape == (16, 3)
    assert mb["dones"].shape == (16,)
    # transitions are internally consistent: next_obs == obs + 1 elementwise
    np.testing.assert_allclose(mb["next_observations"], mb["observations"] + 1.0)
    # rewards come from the stored set
    assert set(np.unique(mb["rewards"])).issubset(set(float(v) for v in range(20)))


def test_sampling_with_replacement_distribution():
    rb = ReplayBuffer(buffer_size=1000)
    rb.add_experience(make_experience(10))
    mb = rb.sample_minibatch(1000)
    # with replacement over 10 items, every item should appear
    assert len(np.unique(mb["rewards"])) == 10


def test_next_observation_semantics_across_episodes():
    """Last transition of an episode pairs with last_observation."""
    rb = ReplayBuffer(buffer_size=100)
    rb.add_experience(make_experience(5))
    obs = rb._storage["observations"][:5].numpy()
    nxt = rb._storage["next_observations"][:5].numpy()
    np.testing.assert_allclose(nxt[-1], np.full(3, 5.0))
    np.testing.assert_allclose(nxt[:-1], obs[1:])
