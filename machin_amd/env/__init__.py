from . import envs, utils, wrappers

__all__ = ["envs", "utils", "wrappers"]
