"""Dataset abstractions for the auto layer.

Parity target: reference ``machin/auto/dataset.py`` — ``RLDataset``
(:75) yielding one interaction unit (episode) per ``__next__`` and
``DatasetResult`` (:94) carrying observations, logged scalars and
logged media; ``determine_precision`` inspecting model dtypes.
"""
from typing import Any, Callable, Dict, List

import torch as t


class DatasetResult:
    def __init__(self, observations: List[Dict[str, Any]] = None,
                 logs: List[Dict[str, Any]] = None):
        self.observations = observations or []
        self.logs = logs or []

    def add_observation(self, obs: Dict[str, Any]):
        self.observations.append(obs)

    def add_log(self, log: Dict[str, Any]):
        self.logs.append(log)

    def __len__(self):
        return len(self.observations)


class RLDataset:
    """Iterable of DatasetResult; one episode per item."""

    early_stopping_monitor = "total_reward"

    def __init__(self, **__):
        pass

    def __iter__(self):
        return self

    def __next__(self) -> DatasetResult:
        raise NotImplementedError


def determine_precision(models) -> t.dtype:
    dtypes = set()
    for model in models:
        for p in model.parameters():
            dtypes.add(p.dtype)
    if len(dtypes) > 1:
        raise RuntimeError(f"Multiple parameter dtypes: {dtypes}")
    return dtypes.pop() if dtypes else t.float32
