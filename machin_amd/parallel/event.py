"""Composable events.

Parity target: reference ``machin/parallel/event.py`` (:7-81):
``OrEvent`` / ``AndEvent`` combining ``threading.Event`` objects by
intercepting their set()/clear() calls.
"""
import threading
from typing import Callable

Event = threading.Event


class MultiEvent:
    """Base for composed events; re-evaluates on every child change."""

    def __init__(self, *events):
        self._events = events
        self._event = threading.Event()
        for e in events:
            self._intercept(e)
        self._update()

    def _intercept(self, event):
        if isinstance(event, MultiEvent):
            event._parents = getattr(event, "_parents", []) + [self]
            return
        orig_set, orig_clear = event.set, event.clear
        parent = self

        def new_set():
            orig_set()
            parent._update()

        def new_clear():
            orig_clear()
            parent._update()

        event.set = new_set
        event.clear = new_clear

    def _child_values(self):
        return [e.is_set() for e in self._events]

    def _combine(self, values) -> bool:
        raise NotImplementedError

    def _update(self):
        if self._combine(self._child_values()):
            self._event.set()
        else:
            self._event.clear()
        for p in getattr(self, "_parents", []):
            p._update()

    def is_set(self) -> bool:
        return self._event.is_set()

    def wait(self, timeout: float = None) -> bool:
        return self._event.wait(timeout)

    def set(self):
        raise RuntimeError("Composed events cannot be set directly.")

    def clear(self):
        raise RuntimeError("Composed events cannot be cleared directly.")


class OrEvent(MultiEvent):
    def _combine(self, values) -> bool:
        return any(values)


class AndEvent(MultiEvent):
    def _combine(self, values) -> bool:
        return all(values)
