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
from .a3c import A3C
from .apex import DDPGApex, DQNApex
from .impala import IMPALA
from .ars import ARS

__all__ = [
    "A3C",
    "DQNApex",
    "DDPGApex",
    "IMPALA",
    "ARS",
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
from .maddpg import MADDPG  # noqa: E402

__all__.append("MADDPG")
