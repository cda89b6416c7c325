from .classic_control import CartPoleEnv, PendulumEnv, Space, make
from .pixel_catch import PixelCatchEnv
from .simple_spread import SimpleSpreadEnv

__all__ = [
    "CartPoleEnv",
    "PendulumEnv",
    "PixelCatchEnv",
    "SimpleSpreadEnv",
    "Space",
    "make",
]
