from .policy import Policy
from .stochastic_policy import StochasticPolicy
from .categorical_policy import CategoricalPolicy
from .gaussian_policy import GaussianPolicy
from .deterministic_policy import DeterministicPolicy
from .random_policy import RandomPolicy

__all__ = [
    "Policy",
    "StochasticPolicy",
    "CategoricalPolicy",
    "GaussianPolicy",
    "DeterministicPolicy",
    "RandomPolicy",
]
