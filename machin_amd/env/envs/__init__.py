from .classic_control import CartPoleEnv, PendulumEnv, Space, make

__all__ = ["CartPoleEnv", "PendulumEnv", "Space", "make"]
from .simple_spread import SimpleSpreadEnv  # noqa: E402

__all__.append("SimpleSpreadEnv")
