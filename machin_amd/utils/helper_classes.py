"""Small helper classes (parity: reference machin/utils/helper_classes.py)."""
import time


class Counter:
    """An integer counter supporting comparison and in-place stepping."""

    def __init__(self, start: int = 0, step: int = 1):
        self._count = start
        self._step = step

    def count(self):
        """Increase the counter by its step size."""
        self._count += self._step
        return self

    def get(self) -> int:
        return self._count

    def reset(self):
        self._count = 0
        return self

    def __eq__(self, other):
        return self._count == int(other)

    def __lt__(self, other):
        return self._count < int(other)

    def __le__(self, other):
        return self._count <= int(other)

    def __gt__(self, other):
        return self._count > int(other)

    def __ge__(self, other):
        return self._count >= int(other)

    def __mod__(self, other):
        return self._count % int(other)

    def __int__(self):
        return self._count

    def __index__(self):
        return self._count

    def __repr__(self):
        return f"Counter({self._count})"


class Switch:
    """A boolean flag with on/off/flip."""

    def __init__(self, state: bool = False):
        self._on = bool(state)

    def get(self) -> bool:
        return self._on

    def on(self):
        self._on = True

    def off(self):
        self._on = False

    def flip(self):
        self._on = not self._on


class Trigger(Switch):
    """A switch that turns itself off once observed in the on state."""

    def get(self) -> bool:
        state = self._on
        if state:
            self._on = False
        return state


class Timer:
    """Wall-clock stopwatch."""

    def __init__(self):
        self._begin = time.monotonic()

    def begin(self):
        self._begin = time.monotonic()

    def end(self) -> float:
        return time.monotonic() - self._begin


class Object:
    """Attribute bag: any attribute may be read/written; missing reads
    return ``None``. Also supports dict-style access and a ``data``
    attribute exposing the underlying dict."""

    def __init__(self, data: dict = None, const_attrs: set = None):
        super().__setattr__("_data", dict(data or {}))
        super().__setattr__("_const", set(const_attrs or ()))

    @property
    def data(self):
        return self._data

    @data.setter
    def data(self, value):
        super().__setattr__("_data", dict(value))

    def attr(self, name, value=None, change: bool = False):
        if change:
            self._data[name] = value
        return self._data.get(name)

    def __getattr__(self, item):
        if item.startswith("_"):
            raise AttributeError(item)
        return self._data.get(item)

    def __setattr__(self, key, value):
        if key == "data":
            object.__setattr__(self, "_data", dict(value))
            return
        if key in self._const:
            raise RuntimeError(f"attribute {key} is constant")
        self._data[key] = value

    def __getitem__(self, item):
        return self._data.get(item)

    def __setitem__(self, key, value):
        self.__setattr__(key, value)

    def __contains__(self, item):
        return item in self._data

    def call(self, *args, **kwargs):
        raise NotImplementedError
