"""Network base with declared I/O devices.

Parity target: reference ``machin/model/nets/base.py`` —
``NeuralNetworkModule`` (:7) plus ``static_module_wrapper`` (:108) and
``dynamic_module_wrapper`` (:125): a module that declares which device
its inputs must arrive on and which device its outputs leave on, so the
framework's ``safe_call`` can route tensors without user bookkeeping.
"""
from typing import Union

import torch as t
import torch.nn as nn


class NeuralNetworkModule(nn.Module):
    """``nn.Module`` with declared ``input_device`` / ``output_device``.

    Set them directly, or point ``input_module`` / ``output_module`` at
    sub-modules whose parameter placement determines them dynamically.
    """

    def __init__(self):
        super().__init__()
        self.input_module: nn.Module = None
        self.output_module: nn.Module = None
        self._input_device = None
        self._output_device = None

    def set_input_module(self, module: nn.Module):
        self.input_module = module

    def set_output_module(self, module: nn.Module):
        self.output_module = module

    @staticmethod
    def find_child(module: nn.Module, find_first: bool = True):
        """Descend into containers to the first/last real layer."""
        while isinstance(module, (nn.Sequential, nn.ModuleList)):
            if len(module) == 0:
                raise RuntimeError("Empty container module.")
            module = module[0] if find_first else module[-1]
        return module

    @staticmethod
    def _device_of(module: nn.Module):
        for p in module.parameters():
            return p.device
        for b in module.buffers():
            return b.device
        return None

    @property
    def input_device(self) -> Union[t.device, None]:
        if self._input_device is not None:
            return self._input_device
        if self.input_module is not None:
            return self._device_of(self.find_child(self.input_module, True))
        return self._device_of(self)

    @input_device.setter
    def input_device(self, device):
        self._input_device = t.device(device) if device is not None else None

    @property
    def output_device(self) -> Union[t.device, None]:
        if self._output_device is not None:
            return self._output_device
        if self.output_module is not None:
            return self._device_of(self.find_child(self.output_module, False))
        return self._device_of(self)

    @output_device.setter
    def output_device(self, device):
        self._output_device = t.device(device) if device is not None else None


def static_module_wrapper(
    wrapped_module: nn.Module,
    input_device: Union[str, t.device],
    output_device: Union[str, t.device],
) -> nn.Module:
    """Attach fixed input/output devices to an ordinary module."""
    wrapped_module.input_device = t.device(input_device)
    wrapped_module.output_device = t.device(output_device)
    return wrapped_module


def dynamic_module_wrapper(wrapped_module: nn.Module) -> NeuralNetworkModule:
    """Wrap a module; I/O devices follow its parameter placement."""
    wrapper = NeuralNetworkModule()
    wrapper.add_module("wrapped_module", wrapped_module)
    wrapper.set_input_module(wrapped_module)
    wrapper.set_output_module(wrapped_module)
    wrapper.forward = wrapped_module.forward
    return wrapper
