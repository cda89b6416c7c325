"""Shared algorithm utilities.

Parity target: reference ``machin/frame/algorithms/utils.py`` —
``soft_update`` (:8), ``hard_update`` (:30), ``safe_call`` (:52),
``safe_return``, config resolvers (:206-312), ``FakeOptimizer`` (:315).

MI355X note: ``soft_update`` is a fused multi-tensor polyak update — on
a ROCm device it runs through ``machin_amd.ops.polyak_update_`` (one HIP
kernel over all parameter chunks, one HBM pass) instead of a python
per-parameter loop; elsewhere it falls back to ``torch._foreach_*``.
"""
import inspect
from typing import Any, Callable, Dict, Union

import torch as t
import torch.nn as nn


# ----------------------------------------------------------------------
# target-network updates
# ----------------------------------------------------------------------
def soft_update(target_net: nn.Module, source_net: nn.Module, update_rate: float):
    """Polyak: target = target·(1−τ) + source·τ over every parameter.

    On GPU a :class:`machin_amd.ops.FusedPolyak` plan is cached on the
    target net: one kernel launch per call with a precomputed device
    pointer table (beats both the per-param reference loop,
    machin/frame/algorithms/utils.py:8-27, and per-call table builds).
    Set MACHIN_AMD_FUSED_POLYAK=0 to force torch _foreach.
    """
    import os

    tgt = [p.data for p in target_net.parameters()]
    src = [p.data for p in source_net.parameters()]
    if not tgt:
        return
    if (
        tgt[0].is_cuda
        and os.environ.get("MACHIN_AMD_FUSED_POLYAK", "1") != "0"
    ):
        from ... import ops

        if ops.available():
            plan = target_net.__dict__.get("_machin_polyak_plan")
            if plan is None or not plan.matches(tgt, src):
                try:
                    plan = ops.FusedPolyak(tgt, src)
                    target_net.__dict__["_machin_polyak_plan"] = plan
                except ValueError:
                    plan = None
            if plan is not None:
                plan(update_rate)
                return
    t._foreach_mul_(tgt, 1.0 - update_rate)
    t._foreach_add_(tgt, src, alpha=update_rate)


def hard_update(target_net: nn.Module, source_net: nn.Module):
    """Copy every parameter of ``source_net`` into ``target_net``."""
    # unwrap DDP containers: a wrapped source's state_dict carries
    # "module."-prefixed keys that would not match a bare target
    target_net = _unwrap(target_net)
    source_net = _unwrap(source_net)
    target_net.load_state_dict(source_net.state_dict())


# ----------------------------------------------------------------------
# model invocation with automatic tensor routing
# ----------------------------------------------------------------------
def _unwrap(model: nn.Module) -> nn.Module:
    # our DDP wrapper and torch containers expose .module
    inner = getattr(model, "module", None)
    return inner if isinstance(inner, nn.Module) else model


def safe_call(model: nn.Module, *named_args: Dict[str, Any], method: str = None):
    """Call ``model`` (or ``model.method``) with keyword arguments merged
    from the given dicts, filtered to the callable's signature, with
    every tensor argument moved to the model's ``input_device``.

    Returns the model's output unchanged.
    """
    inner = _unwrap(model)
    func = getattr(inner, method) if method else None
    if func is None:
        func = inner.forward
        call_target = model  # keep hooks / DDP wrapping on plain forward
    else:
        call_target = None

    input_device = getattr(inner, "input_device", None)
    if input_device is None:
        for p in inner.parameters():
            input_device = p.device
            break

    sig = inspect.signature(func)
    params = sig.parameters
    accepts_kwargs = any(
        p.kind == inspect.Parameter.VAR_KEYWORD for p in params.values()
    )
    required = {
        name
        for name, p in params.items()
        if p.default is inspect.Parameter.empty
        and p.kind
        in (inspect.Parameter.POSITIONAL_OR_KEYWORD, inspect.Parameter.KEYWORD_ONLY)
        and name != "self"
    }

    kwargs = {}
    for d in named_args:
        if d is None:
            continue
        for k, v in d.items():
            if accepts_kwargs or k in params:
                kwargs[k] = v

    missing = required - set(kwargs.keys())
    if missing:
        raise RuntimeError(
            f"Model {type(inner).__name__}.{method or 'forward'} requires "
            f"arguments {sorted(missing)} which were not provided. "
            f"Provided keys: {sorted(kwargs.keys())}."
        )

    if input_device is not None:
        for k, v in kwargs.items():
            if t.is_tensor(v) and v.device != input_device:
                kwargs[k] = v.to(input_device, non_blocking=True)

    if call_target is not None:
        return call_target(**kwargs)
    return func(**kwargs)


def safe_return(result):
    """Unwrap a 1-tuple model output."""
    if isinstance(result, tuple) and len(result) == 1:
        return result[0]
    return result


# ----------------------------------------------------------------------
# config resolvers
# ----------------------------------------------------------------------
def _resolve_class(spec: Union[str, type], search_modules) -> type:
    if isinstance(spec, type):
        return spec
    if callable(spec) and not isinstance(spec, str):
        return spec
    if "." in spec:
        import importlib

        mod_name, _, cls_name = spec.rpartition(".")
        mod = importlib.import_module(mod_name)
        return getattr(mod, cls_name)
    for mod in search_modules:
        if hasattr(mod, spec):
            return getattr(mod, spec)
    import __main__

    if hasattr(__main__, spec):
        return getattr(__main__, spec)
    raise ValueError(f"Cannot resolve class {spec!r}.")


def assert_and_get_valid_models(models):
    """Resolve a list of model classes from classes or name strings."""
    import __main__

    out = []
    for m in models:
        out.append(_resolve_class(m, [__main__]))
    return out


def assert_and_get_valid_optimizer(optimizer):
    return _resolve_class(optimizer, [t.optim])


def assert_and_get_valid_criterion(criterion):
    return _resolve_class(criterion, [nn, nn.functional])


def assert_and_get_valid_lr_scheduler(lr_scheduler):
    if lr_scheduler is None:
        return None
    return _resolve_class(lr_scheduler, [t.optim.lr_scheduler])


class FakeOptimizer(t.optim.Optimizer):
    """An optimizer that does nothing (A3C local step is a no-op: real
    stepping happens on the gradient server)."""

    def __init__(self, params, *_, **__):
        super().__init__(params, {})

    def zero_grad(self, set_to_none: bool = True):
        super().zero_grad(set_to_none=set_to_none)

    def step(self, closure=None):
        if closure is not None:
            with t.enable_grad():
                return closure()
        return None
