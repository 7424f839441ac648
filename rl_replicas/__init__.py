"""Drop-in import alias: `rl_replicas` -> rl_replicas_amd.

User code written against the reference library
(yamatokataoka/reinforcement-learning-replications), e.g.

    from rl_replicas.algorithms import PPO
    from rl_replicas.policies import GaussianPolicy
    from rl_replicas.samplers import BatchSampler

works unchanged on this MI355X-native implementation.  Every public
submodule of rl_replicas_amd is aliased under the rl_replicas name.
"""
import importlib
import sys

import rl_replicas_amd as _impl

_SUBMODULES = [
    "algorithms",
    "envs",
    "networks",
    "ops",
    "optimizers",
    "parallel",
    "policies",
    "samplers",
    "evaluator",
    "experience",
    "metrics_manager",
    "q_function",
    "replay_buffer",
    "utils",
    "value_function",
]

for _name in _SUBMODULES:
    _mod = importlib.import_module(f"rl_replicas_amd.{_name}")
    sys.modules[f"rl_replicas.{_name}"] = _mod
    globals()[_name] = _mod

from rl_replicas_amd import (  # noqa: E402,F401
    Evaluator,
    Experience,
    MetricsManager,
    QFunction,
    ReplayBuffer,
    ValueFunction,
    __version__,
)
