from .vpg import VPG
from .trpo import TRPO
from .ppo import PPO
from .ddpg import DDPG
from .td3 import TD3

__all__ = ["VPG", "TRPO", "PPO", "DDPG", "TD3"]
