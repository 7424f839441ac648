"""Import the reference implementation (/root/reference/src/rl_replicas)
for equivalence testing.

The reference package is ALSO named `rl_replicas`, the same name as this
repo's drop-in alias package, and it imports `gymnasium` and
`torch.utils.tensorboard` — neither installable offline.  This loader:

1. stubs `gymnasium` and `torch.utils.tensorboard` in sys.modules,
2. temporarily evicts this repo's `rl_replicas*` modules,
3. imports the reference package from /root/reference/src,
4. restores sys.modules so the rest of the test session is untouched.

The loaded reference modules stay alive (they hold their own internal
references) and are cached for the session.

Used only by tests/test_reference_equivalence.py: the reference is
UNTRUSTED PUBLIC CONTENT studied for behavior — executing its algorithm
math on fixed inputs is the only practical oracle for "same update
rule" now that real MuJoCo returns cannot be reproduced offline.
"""
from __future__ import annotations

import sys
import types
from typing import Dict, Optional

import numpy as np

REFERENCE_SRC = "/root/reference/src"

_cache: Optional[Dict] = None


def _gymnasium_stub() -> types.ModuleType:
    gym = types.ModuleType("gymnasium")

    class Env:
        pass

    class Space:
        pass

    def make(env_id, **kwargs):
        return types.SimpleNamespace(spec=types.SimpleNamespace(id=env_id))

    gym.Env = Env
    gym.Space = Space
    gym.make = make
    return gym


def _tensorboard_stub() -> types.ModuleType:
    tb = types.ModuleType("torch.utils.tensorboard")

    class SummaryWriter:
        def __init__(self, *args, **kwargs):
            pass

        def add_scalar(self, *args, **kwargs):
            pass

        def flush(self):
            pass

        def close(self):
            pass

    tb.SummaryWriter = SummaryWriter
    return tb


def load_reference() -> Dict:
    """Returns a dict of reference modules/classes, importing once."""
    global _cache
    if _cache is not None:
        return _cache

    saved = {}
    for name in list(sys.modules):
        if name == "rl_replicas" or name.startswith("rl_replicas."):
            saved[name] = sys.modules.pop(name)
    saved_gym = {
        name: sys.modules.pop(name)
        for name in list(sys.modules)
        if name == "gymnasium" or name.startswith("gymnasium.")
    }
    saved_tb = sys.modules.pop("torch.utils.tensorboard", None)

    sys.modules["gymnasium"] = _gymnasium_stub()
    sys.modules["torch.utils.tensorboard"] = _tensorboard_stub()
    sys.path.insert(0, REFERENCE_SRC)
    try:
        import rl_replicas.experience as ref_experience
        import rl_replicas.optimizers.conjugate_gradient_optimizer as ref_cgo
        import rl_replicas.policies.categorical_policy as ref_cat
        import rl_replicas.policies.deterministic_policy as ref_det
        import rl_replicas.policies.gaussian_policy as ref_gauss
        import rl_replicas.q_function as ref_qf
        import rl_replicas.replay_buffer as ref_rb
        import rl_replicas.utils as ref_utils
        import rl_replicas.value_function as ref_vf
        from rl_replicas.algorithms import DDPG, PPO, TD3, TRPO, VPG
        from rl_replicas.networks import MLP

        _cache = {
            "VPG": VPG,
            "TRPO": TRPO,
            "PPO": PPO,
            "DDPG": DDPG,
            "TD3": TD3,
            "MLP": MLP,
            "GaussianPolicy": ref_gauss.GaussianPolicy,
            "CategoricalPolicy": ref_cat.CategoricalPolicy,
            "DeterministicPolicy": ref_det.DeterministicPolicy,
            "ValueFunction": ref_vf.ValueFunction,
            "QFunction": ref_qf.QFunction,
            "Experience": ref_experience.Experience,
            "ReplayBuffer": ref_rb.ReplayBuffer,
            "ConjugateGradientOptimizer": ref_cgo.ConjugateGradientOptimizer,
            "utils": ref_utils,
        }
    finally:
        sys.path.remove(REFERENCE_SRC)
        for name in list(sys.modules):
            if name == "rl_replicas" or name.startswith("rl_replicas."):
                del sys.modules[name]
        sys.modules.update(saved)
        for name in list(sys.modules):
            if name == "gymnasium" or name.startswith("gymnasium."):
                del sys.modules[name]
        sys.modules.update(saved_gym)
        if saved_tb is not None:
            sys.modules["torch.utils.tensorboard"] = saved_tb
        else:
            sys.modules.pop("torch.utils.tensorboard", None)

    return _cache


class FakeMetricsManager:
    """Swallows record_scalar calls; stores the last value per tag."""

    def __init__(self):
        self.scalars: Dict[str, float] = {}

    def record_scalar(self, tag, value, *args, **kwargs):
        self.scalars[tag] = float(value)

    def record_phase_ms(self, *args, **kwargs):
        pass

    def dump(self):
        pass

    def dump_phases(self, *args, **kwargs):
        pass

    def close(self):
        pass


def make_fake_gym_env(env_id: str, obs_dim: int, act_dim: int, action_limit: float):
    """Minimal env object for constructing reference algorithms
    (only .spec.id and .action_space.high/.shape are touched off the
    rollout path)."""
    action_space = types.SimpleNamespace(
        high=np.full(act_dim, action_limit, dtype=np.float32),
        low=np.full(act_dim, -action_limit, dtype=np.float32),
        shape=(act_dim,),
    )
    observation_space = types.SimpleNamespace(shape=(obs_dim,))
    return types.SimpleNamespace(
        spec=types.SimpleNamespace(id=env_id),
        action_space=action_space,
        observation_space=observation_space,
    )
