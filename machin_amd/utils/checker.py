"""Model I/O and parameter checkers.

Parity target: reference ``machin/utils/checker.py`` (:10-130+):
forward/backward hooks that validate input/output/param tensors
(NaN/Inf, custom shape checks) and optionally log histograms to
tensorboard. ``check_model`` returns a cancel function.
"""
from typing import Callable, List

import torch as t
import torch.nn as nn

from .logging import default_logger


class CheckError(Exception):
    pass


def check_nan(tensor: t.Tensor, name: str = ""):
    if t.is_tensor(tensor) and tensor.is_floating_point():
        if t.isnan(tensor).any():
            raise CheckError(f"NaN detected in {name}")


def check_inf(tensor: t.Tensor, name: str = ""):
    if t.is_tensor(tensor) and tensor.is_floating_point():
        if t.isinf(tensor).any():
            raise CheckError(f"Inf detected in {name}")


def check_shape(tensor: t.Tensor, required_shape: List[int], name: str = ""):
    shape = list(tensor.shape)
    if shape != list(required_shape):
        raise CheckError(
            f"Shape mismatch in {name}: expected {required_shape}, "
            f"got {shape}"
        )


def _walk(value, fn, name):
    if t.is_tensor(value):
        fn(value, name)
    elif isinstance(value, (list, tuple)):
        for i, v in enumerate(value):
            _walk(v, fn, f"{name}[{i}]")
    elif isinstance(value, dict):
        for k, v in value.items():
            _walk(v, fn, f"{name}[{k}]")


def check_model(
    writer,
    model: nn.Module,
    input_check_hooks: List[Callable] = (check_nan, check_inf),
    output_check_hooks: List[Callable] = (check_nan, check_inf),
    param_check_hooks: List[Callable] = (check_nan, check_inf),
    input_check_interval: int = 1,
    output_check_interval: int = 1,
    param_check_interval: int = 100,
    name: str = "",
) -> Callable[[], None]:
    """Install check hooks on every sub-module of ``model``.

    ``writer`` is a tensorboard-like object with ``add_histogram``
    (or None). Returns a function that removes all hooks."""
    handles = []
    counters = {"in": 0, "out": 0, "param": 0}

    def fwd_pre(module, inputs):
        counters["in"] += 1
        if counters["in"] % input_check_interval:
            return
        for hook in input_check_hooks:
            _walk(inputs, hook, f"{name}/{type(module).__name__}/input")

    def fwd_post(module, inputs, output):
        counters["out"] += 1
        if counters["out"] % output_check_interval:
            return
        for hook in output_check_hooks:
            _walk(output, hook, f"{name}/{type(module).__name__}/output")
        counters["param"] += 1
        if counters["param"] % param_check_interval == 0:
            for pname, p in module.named_parameters(recurse=False):
                full = f"{name}/{type(module).__name__}/{pname}"
                for hook in param_check_hooks:
                    hook(p, full)
                if writer is not None:
                    try:
                        writer.add_histogram(full, p.detach().cpu())
                    except Exception:  # noqa: BLE001 - writer optional
                        default_logger.warning(
                            f"Failed to write histogram for {full}"
                        )

    for module in model.modules():
        handles.append(module.register_forward_pre_hook(fwd_pre))
        handles.append(module.register_forward_hook(fwd_post))

    def cancel():
        for h in handles:
            h.remove()

    return cancel
