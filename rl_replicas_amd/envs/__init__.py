from .spaces import Box, Discrete, Space
from .core import Env, EnvSpec, make, register, registered_ids
from .classic import CartPoleEnv, PendulumEnv
from .synthetic import MUJOCO_SHAPES, SyntheticEnv
from .device import DevicePendulumEnv, DeviceVectorEnv
from .subproc import SubprocVectorEnv
from .vector import SerialVectorEnv, VectorEnv

__all__ = [
    "Box",
    "Discrete",
    "Space",
    "Env",
    "EnvSpec",
    "make",
    "register",
    "registered_ids",
    "CartPoleEnv",
    "PendulumEnv",
    "SyntheticEnv",
    "MUJOCO_SHAPES",
    "VectorEnv",
    "SerialVectorEnv",
    "SubprocVectorEnv",
    "DeviceVectorEnv",
    "DevicePendulumEnv",
]
