from .helper_classes import Counter, Object, Switch, Timer, Trigger
from .logging import default_logger, fake_logger

__all__ = [
    "Counter",
    "Object",
    "Switch",
    "Timer",
    "Trigger",
    "default_logger",
    "fake_logger",
]
