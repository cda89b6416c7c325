"""Profiling ranges for rocprof timelines.

SURVEY.md §5.1: the reference has no tracing; machin_amd emits
roctx-compatible ranges (torch.cuda.nvtx maps to rocTX on ROCm) so
`rocprofv3 --marker-trace` attributes GPU time to framework phases.
No-ops when CUDA is unavailable.
"""
import contextlib
import functools

import torch as t

_enabled = t.cuda.is_available()


def set_enabled(flag: bool):
    global _enabled
    _enabled = bool(flag) and t.cuda.is_available()


@contextlib.contextmanager
def trace_range(name: str):
    """Context manager emitting a rocTX range."""
    if _enabled:
        t.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            t.cuda.nvtx.range_pop()
    else:
        yield


def traced(name: str = None):
    """Decorator form of :func:`trace_range`."""

    def deco(fn):
        label = name or fn.__qualname__

        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            with trace_range(label):
                return fn(*args, **kwargs)

        return wrapper

    return deco
