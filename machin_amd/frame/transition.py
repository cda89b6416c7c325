"""Transition records.

Parity target: reference ``machin/frame/transition.py`` (TransitionBase
at :9, Transition at :224) — same three-group attribute contract:

* **major attributes**: ``Dict[str, torch.Tensor]`` (e.g. state, action,
  next_state); on sampling each inner key is concatenated along dim 0.
* **sub attributes**: scalar or ``torch.Tensor`` (e.g. reward, terminal).
* **custom attributes**: arbitrary objects, passed through untouched.

This implementation is written for the MI355X data path: transitions are
lightweight views (no deepcopy on construction); copying / device moves
happen once, inside the storage layer, so a host actor can hand the same
dict to ``store_episode`` without paying python-object churn twice.
"""
from typing import Any, Dict, Iterable, List, Union

import torch as t

Scalar = Union[int, float, bool]


class TransitionBase:
    """Base transition with user-declared attribute groups."""

    _reserved = {
        "_major_attr",
        "_sub_attr",
        "_custom_attr",
        "_batch_size",
        "_inited",
    }

    def __init__(
        self,
        major_attr: Iterable[str],
        sub_attr: Iterable[str],
        custom_attr: Iterable[str],
        major_data: Iterable[Dict[str, t.Tensor]],
        sub_data: Iterable[Union[Scalar, t.Tensor]],
        custom_data: Iterable[Any],
    ):
        object.__setattr__(self, "_inited", False)
        self._major_attr = list(major_attr)
        self._sub_attr = list(sub_attr)
        self._custom_attr = list(custom_attr)
        for attr, data in zip(self._major_attr, major_data):
            setattr(self, attr, data)
        for attr, data in zip(self._sub_attr, sub_data):
            setattr(self, attr, data)
        for attr, data in zip(self._custom_attr, custom_data):
            setattr(self, attr, data)
        object.__setattr__(self, "_inited", True)
        self._batch_size = self._detect_batch_size()
        self._check_validity()

    # ------------------------------------------------------------------
    @property
    def major_attr(self) -> List[str]:
        return self._major_attr

    @property
    def sub_attr(self) -> List[str]:
        return self._sub_attr

    @property
    def custom_attr(self) -> List[str]:
        return self._custom_attr

    def keys(self) -> List[str]:
        return self._major_attr + self._sub_attr + self._custom_attr

    def items(self):
        for k in self.keys():
            yield k, getattr(self, k)

    def has_keys(self, keys: Iterable[str]) -> bool:
        return all(k in self.keys() for k in keys)

    def __len__(self):
        return len(self.keys())

    def __getitem__(self, item):
        return getattr(self, item)

    def __setitem__(self, key, value):
        if key not in self.keys():
            raise RuntimeError(
                f"You cannot dynamically set new attribute {key!r} on a "
                f"transition; declare it in the constructor."
            )
        setattr(self, key, value)

    def __setattr__(self, key, value):
        if getattr(self, "_inited", False) and key not in self._reserved:
            if key not in self.keys():
                raise RuntimeError(
                    f"You cannot dynamically set new attribute {key!r} on a "
                    f"transition; declare it in the constructor."
                )
        object.__setattr__(self, key, value)

    # ------------------------------------------------------------------
    def to(self, device: Union[str, t.device]):
        """Move every tensor attribute to ``device`` in-place; returns self."""
        for ma in self._major_attr:
            d = getattr(self, ma)
            for k, v in d.items():
                d[k] = v.to(device)
        for sa in self._sub_attr:
            v = getattr(self, sa)
            if t.is_tensor(v):
                object.__setattr__(self, sa, v.to(device))
        return self

    def _detach(self):
        """Detach every tensor attribute from the autograd graph; returns self."""
        for ma in self._major_attr:
            d = getattr(self, ma)
            for k, v in d.items():
                d[k] = v.detach()
        for sa in self._sub_attr:
            v = getattr(self, sa)
            if t.is_tensor(v):
                object.__setattr__(self, sa, v.detach())
        return self

    def clone(self):
        """Deep-copy tensors (clone) and shallow-copy custom attrs."""
        major = [
            {k: v.clone() for k, v in getattr(self, ma).items()}
            for ma in self._major_attr
        ]
        sub = [
            getattr(self, sa).clone() if t.is_tensor(getattr(self, sa))
            else getattr(self, sa)
            for sa in self._sub_attr
        ]
        custom = [getattr(self, ca) for ca in self._custom_attr]
        return type(self)._construct(
            self._major_attr, self._sub_attr, self._custom_attr, major, sub, custom
        )

    @classmethod
    def _construct(cls, major_attr, sub_attr, custom_attr, major, sub, custom):
        obj = TransitionBase.__new__(cls)
        TransitionBase.__init__(
            obj, major_attr, sub_attr, custom_attr, major, sub, custom
        )
        return obj

    # ------------------------------------------------------------------
    @property
    def batch_size(self) -> int:
        return self._batch_size

    def _detect_batch_size(self) -> int:
        for ma in self._major_attr:
            d = getattr(self, ma)
            if not isinstance(d, dict):
                continue
            for v in d.values():
                if t.is_tensor(v) and v.dim() >= 1:
                    return v.shape[0]
        return 1

    def _check_validity(self):
        bs = self._batch_size
        for ma in self._major_attr:
            d = getattr(self, ma)
            if not isinstance(d, dict):
                raise ValueError(
                    f"Major attribute {ma!r} must be a dict of tensors, "
                    f"got {type(d).__name__}."
                )
            for k, v in d.items():
                if not t.is_tensor(v):
                    raise ValueError(
                        f"Key {k!r} of major attribute {ma!r} must be a "
                        f"torch.Tensor, got {type(v).__name__}."
                    )
                if v.dim() < 1 or v.shape[0] != bs:
                    raise ValueError(
                        f"Key {k!r} of major attribute {ma!r} has batch size "
                        f"{tuple(v.shape)[:1]}, expected {bs}."
                    )
        for sa in self._sub_attr:
            v = getattr(self, sa)
            if t.is_tensor(v):
                if v.dim() < 1 or v.shape[0] != bs:
                    raise ValueError(
                        f"Sub attribute {sa!r} has batch size "
                        f"{tuple(v.shape)[:1]}, expected {bs}."
                    )
            elif not isinstance(v, (int, float, bool)):
                raise ValueError(
                    f"Sub attribute {sa!r} must be a scalar or tensor, got "
                    f"{type(v).__name__}."
                )


class Transition(TransitionBase):
    """The default RL transition: state / action / next_state (major),
    reward / terminal (sub), plus any user keyword args as custom attrs.
    Batch size is fixed to 1 (one environment step)."""

    state: Dict[str, t.Tensor]
    action: Dict[str, t.Tensor]
    next_state: Dict[str, t.Tensor]
    reward: Union[float, t.Tensor]
    terminal: Union[bool, t.Tensor]

    def __init__(
        self,
        state: Dict[str, t.Tensor],
        action: Dict[str, t.Tensor],
        next_state: Dict[str, t.Tensor],
        reward: Union[float, t.Tensor],
        terminal: Union[bool, t.Tensor],
        **kwargs,
    ):
        custom_keys = list(kwargs.keys())
        super().__init__(
            major_attr=["state", "action", "next_state"],
            sub_attr=["reward", "terminal"],
            custom_attr=custom_keys,
            major_data=[state, action, next_state],
            sub_data=[reward, terminal],
            custom_data=[kwargs[k] for k in custom_keys],
        )

    def _check_validity(self):
        if self._batch_size != 1:
            raise ValueError(
                f"Transition batch size must be 1, got {self._batch_size}."
            )
        super()._check_validity()
