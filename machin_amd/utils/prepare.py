"""Checkpoint discovery and loading.

Parity target: reference ``machin/utils/prepare.py`` —
``prep_create_dirs``, ``prep_load_state_dict`` (:40), ``prep_load_model``
(:52-118): finds the newest checkpoint version common to every model by
scanning ``{name}_{version}.pt`` filenames, loads state dicts with
device remapping.
"""
import os
import re
from typing import Dict, Iterable

import torch as t
import torch.nn as nn


def prep_create_dirs(dirs: Iterable[str]):
    for d in dirs:
        os.makedirs(d, exist_ok=True)


def prep_load_state_dict(model: nn.Module, state_dict):
    """Load a state dict, remapping each tensor to the device of the
    model's existing parameter of the same name."""
    model = getattr(model, "module", model)
    own = model.state_dict()
    remapped = {}
    for k, v in state_dict.items():
        if k in own and t.is_tensor(v):
            remapped[k] = v.to(own[k].device)
        else:
            remapped[k] = v
    model.load_state_dict(remapped)


def prep_load_model(
    model_dir: str,
    model_map: Dict[str, nn.Module],
    version: int = -1,
    quiet: bool = False,
):
    """Load every model in ``model_map`` ({save-name: module}) from
    ``model_dir``. ``version=-1`` selects the largest version present
    for ALL names; otherwise the exact version is required."""
    if not os.path.isdir(model_dir):
        raise RuntimeError(f"Model directory {model_dir!r} does not exist.")
    versions = None
    for name in model_map:
        found = set()
        for f in os.listdir(model_dir):
            m = re.fullmatch(re.escape(name) + r"_(\d+)\.pt", f)
            if m:
                found.add(int(m.group(1)))
        versions = found if versions is None else (versions & found)
    if not versions:
        raise RuntimeError(
            f"No common checkpoint version found in {model_dir!r} for "
            f"models {sorted(model_map)}."
        )
    if version == -1:
        version = max(versions)
    elif version not in versions:
        raise RuntimeError(
            f"Version {version} not available for all models; "
            f"common versions: {sorted(versions)}."
        )
    for name, model in model_map.items():
        path = os.path.join(model_dir, f"{name}_{version}.pt")
        state = t.load(path, map_location="cpu", weights_only=True)
        prep_load_state_dict(model, state)
    return version
