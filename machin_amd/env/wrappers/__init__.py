from .base import ParallelWrapperBase
from .parallel import ParallelWrapperDummy, ParallelWrapperSubProc

__all__ = [
    "ParallelWrapperBase",
    "ParallelWrapperDummy",
    "ParallelWrapperSubProc",
]
