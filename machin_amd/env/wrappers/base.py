"""Parallel environment wrapper interface.

Parity target: reference ``machin/env/wrappers/base.py`` (:5-105):
abstract batched-environment API — reset/step/seed/render/close over a
set of sub-environments addressed by index, plus ``active()`` and
``size()`` and space accessors.
"""
from typing import Any, List


class ParallelWrapperBase:
    """Abstract batched environment."""

    def __init__(self, *_, **__):
        pass

    def reset(self, idx: Any = None) -> Any:
        """Reset all (idx=None) or selected sub-environments."""
        raise NotImplementedError

    def step(self, action: Any, idx: Any = None) -> Any:
        """Step selected sub-environments with per-env actions."""
        raise NotImplementedError

    def seed(self, seed: Any = None) -> List[int]:
        raise NotImplementedError

    def render(self, idx: Any = None, *args, **kwargs) -> Any:
        raise NotImplementedError

    def close(self) -> None:
        raise NotImplementedError

    def active(self) -> List[int]:
        """Indexes of environments still running their episode."""
        raise NotImplementedError

    def size(self) -> int:
        raise NotImplementedError

    @property
    def action_space(self) -> Any:
        raise NotImplementedError

    @property
    def observation_space(self) -> Any:
        raise NotImplementedError
