"""rl_replicas_amd: a brand-new MI355X-native RL training library.

Capabilities and public API of `rl_replicas`
(yamatokataoka/reinforcement-learning-replications), built from scratch
for AMD Instinct MI355X (gfx950, CDNA4): PyTorch-ROCm object model,
hand-written HIP kernels for every hot primitive, RCCL-over-xGMI data
parallelism.  See SURVEY.md for the blueprint.
"""
import logging

from rl_replicas_amd.version import __version__

logging.getLogger(__name__).addHandler(logging.NullHandler())

from rl_replicas_amd import algorithms, envs, networks, ops, optimizers, parallel, policies, samplers
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.experience import Experience
from rl_replicas_amd.metrics_manager import MetricsManager
from rl_replicas_amd.q_function import QFunction
from rl_replicas_amd.replay_buffer import ReplayBuffer
from rl_replicas_amd.value_function import ValueFunction

__all__ = [
    "__version__",
    "algorithms",
    "envs",
    "networks",
    "ops",
    "optimizers",
    "parallel",
    "policies",
    "samplers",
    "Evaluator",
    "Experience",
    "MetricsManager",
    "QFunction",
    "ReplayBuffer",
    "ValueFunction",
]
