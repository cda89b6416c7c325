"""Finalization utilities.

Parity target: reference ``machin/parallel/util.py`` (:12):
``Finalize`` — registered cleanup callbacks with priority ordering,
runnable once, also invoked at interpreter exit.
"""
import atexit
import itertools
import weakref

_registry = {}
_counter = itertools.count()


class Finalize:
    """Register a callback to run when ``obj`` dies, when called
    directly, or at interpreter exit; higher ``exitpriority`` runs
    earlier at exit."""

    def __init__(self, obj, callback, args=(), kwargs=None,
                 exitpriority=None):
        self._callback = callback
        self._args = args
        self._kwargs = kwargs or {}
        self._key = (exitpriority, next(_counter))
        self._weakref = (
            weakref.ref(obj, self) if obj is not None else None
        )
        if exitpriority is not None:
            _registry[self._key] = self

    def __call__(self, wr=None):
        if self._callback is None:
            return None
        cb, self._callback = self._callback, None
        _registry.pop(self._key, None)
        return cb(*self._args, **self._kwargs)

    def cancel(self):
        self._callback = None
        _registry.pop(self._key, None)

    def still_active(self) -> bool:
        return self._callback is not None


def _run_finalizers():
    for key in sorted(
        list(_registry.keys()),
        key=lambda k: (-(k[0] if k[0] is not None else 0), k[1]),
    ):
        finalizer = _registry.get(key)
        if finalizer is not None:
            try:
                finalizer()
            except Exception:  # noqa: BLE001 - exit path
                pass


atexit.register(_run_finalizers)
