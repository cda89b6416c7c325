from .trpo import TRPOActorContinuous, TRPOActorDiscrete

__all__ = ["TRPOActorDiscrete", "TRPOActorContinuous"]
