"""Algorithm framework base.

Parity target: reference ``machin/frame/algorithms/base.py``
(TorchFramework :11): ``_is_top`` / ``_is_restorable`` model
registries, versioned ``save/load`` with ``{name}_{version}.pt``
filenames, pluggable backward function, ``generate_config`` /
``init_from_config`` contract, ``is_distributed`` classmethod.
"""
import os
import re
from typing import Any, Callable, Dict, Union

import torch as t

from ...utils.conf import Config
from ...utils.prepare import prep_load_model


class TorchFramework:
    """Base class of all algorithm frameworks."""

    _is_top = []          # model attribute names exposed to servers/DDP
    _is_restorable = []   # model attribute names included in checkpoints

    def __init__(self):
        self._visualized = set()
        self._backward = t.autograd.backward

    # -- registries ----------------------------------------------------
    @property
    def optimizers(self):
        raise NotImplementedError

    @optimizers.setter
    def optimizers(self, optimizers):
        raise NotImplementedError

    @property
    def lr_schedulers(self):
        raise NotImplementedError

    @property
    def top_models(self):
        return [getattr(self, name) for name in self._is_top]

    @property
    def restorable_models(self):
        return [getattr(self, name) for name in self._is_restorable]

    @classmethod
    def get_top_model_names(cls):
        return list(cls._is_top)

    @classmethod
    def get_restorable_model_names(cls):
        return list(cls._is_restorable)

    @classmethod
    def is_distributed(cls) -> bool:
        return False

    # -- backward hook -------------------------------------------------
    @property
    def backward_function(self) -> Callable:
        return self._backward

    def set_backward_function(self, backward_func: Callable):
        if not callable(backward_func):
            raise ValueError("Backward function must be callable.")
        self._backward = backward_func

    # -- multiprocessing -----------------------------------------------
    def enable_multiprocessing(self):
        """Make models share memory so forked workers see updates."""
        for model in self.restorable_models:
            model.share_memory()

    # -- checkpointing -------------------------------------------------
    def save(
        self,
        model_dir: str,
        network_map: Dict[str, str] = None,
        version: int = 0,
    ):
        """Save every restorable model as ``{name}_{version}.pt``."""
        network_map = network_map or {}
        os.makedirs(model_dir, exist_ok=True)
        if version <= 0:
            # auto: next version after the largest on disk
            version = self._find_latest_version(model_dir, network_map) + 1
        for attr in self._is_restorable:
            name = network_map.get(attr, attr)
            model = getattr(self, attr)
            model = getattr(model, "module", model)
            t.save(
                model.state_dict(),
                os.path.join(model_dir, f"{name}_{version}.pt"),
            )
        return version

    def load(
        self,
        model_dir: str,
        network_map: Dict[str, str] = None,
        version: int = -1,
    ):
        """Load restorable models; ``version=-1`` picks the newest
        version common to all of them."""
        network_map = network_map or {}
        prep_load_model(
            model_dir,
            {
                network_map.get(attr, attr): getattr(self, attr)
                for attr in self._is_restorable
            },
            version=version,
        )

    def _find_latest_version(self, model_dir: str, network_map: Dict[str, str]):
        best = 0
        names = {network_map.get(a, a) for a in self._is_restorable}
        if not os.path.isdir(model_dir):
            return best
        for f in os.listdir(model_dir):
            m = re.fullmatch(r"(.+)_(\d+)\.pt", f)
            if m and m.group(1) in names:
                best = max(best, int(m.group(2)))
        return best

    # -- visualization -------------------------------------------------
    def visualize_model(self, final_tensor: t.Tensor, name: str, directory: str):
        """Dump the autograd graph of ``final_tensor`` once per name.
        (torchviz is absent in this image; writes a text rendition.)"""
        if name in self._visualized:
            return
        self._visualized.add(name)
        os.makedirs(directory, exist_ok=True)
        lines = []

        def walk(fn, depth=0):
            if fn is None or depth > 50:
                return
            lines.append("  " * depth + type(fn).__name__)
            for nxt, _ in getattr(fn, "next_functions", ()):
                walk(nxt, depth + 1)

        walk(final_tensor.grad_fn)
        with open(os.path.join(directory, f"{name}.txt"), "w") as f:
            f.write("\n".join(lines))

    # -- config contract -----------------------------------------------
    @classmethod
    def generate_config(cls, config: Union[Dict[str, Any], Config]):
        raise NotImplementedError

    @classmethod
    def init_from_config(
        cls, config: Union[Dict[str, Any], Config], model_device="cpu"
    ):
        raise NotImplementedError
