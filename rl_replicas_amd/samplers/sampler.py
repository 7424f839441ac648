"""Sampler ABC (reference: src/rl_replicas/samplers/sampler.py:7-21)."""
from abc import ABC, abstractmethod

from rl_replicas_amd.experience import Experience
from rl_replicas_amd.policies import Policy


class Sampler(ABC):
    """Collects experience from an environment using a policy."""

    @abstractmethod
    def sample(self, num_samples: int, policy: Policy) -> Experience:
        raise NotImplementedError
