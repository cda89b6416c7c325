from .base import ParallelWrapperBase
from .gym_like import GymAdapter, adapt, adapt_creators, validate_env
from .parallel import ParallelWrapperDummy, ParallelWrapperSubProc

__all__ = [
    "ParallelWrapperBase",
    "ParallelWrapperDummy",
    "ParallelWrapperSubProc",
    "GymAdapter",
    "adapt",
    "adapt_creators",
    "validate_env",
]
