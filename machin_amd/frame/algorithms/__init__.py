from .base import TorchFramework
from .dqn import DQN
from .dqn_per import DQNPer
from .rainbow import RAINBOW
from .ddpg import DDPG
from .hddpg import HDDPG
from .td3 import TD3
from .ddpg_per import DDPGPer
from .sac import SAC
from .a2c import A2C
from .ppo import PPO
from .trpo import TRPO
from .gail import GAIL

__all__ = [
    "TorchFramework",
    "DQN",
    "DQNPer",
    "RAINBOW",
    "DDPG",
    "HDDPG",
    "TD3",
    "DDPGPer",
    "SAC",
    "A2C",
    "PPO",
    "TRPO",
    "GAIL",
]
