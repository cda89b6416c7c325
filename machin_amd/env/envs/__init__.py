from .classic_control import CartPoleEnv, PendulumEnv, Space, make

__all__ = ["CartPoleEnv", "PendulumEnv", "Space", "make"]
