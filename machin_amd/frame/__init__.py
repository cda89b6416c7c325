from .transition import Scalar, Transition, TransitionBase

__all__ = ["Scalar", "Transition", "TransitionBase"]
