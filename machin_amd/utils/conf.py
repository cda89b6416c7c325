"""Config object + JSON / command-line merging.

Parity target: reference ``machin/utils/conf.py`` (:9-131).
"""
import json
from typing import Any, Dict, Union

from .helper_classes import Object


class Config(Object):
    """Attribute-bag configuration: missing keys read as ``None``."""

    def __init__(self, **configs):
        super().__init__(data=configs)


def load_config_cmd(merge_conf: Config = None, args=None) -> Config:
    """Merge ``--conf key=value`` command-line entries into a config.
    Values are parsed as python literals where possible."""
    import argparse
    import ast

    parser = argparse.ArgumentParser(add_help=False)
    parser.add_argument("--conf", action="append", default=[])
    parsed, _ = parser.parse_known_args(args)
    conf = merge_conf if merge_conf is not None else Config()
    for entry in parsed.conf:
        if "=" not in entry:
            raise ValueError(f"Invalid --conf entry {entry!r}, expected key=value.")
        key, _, value = entry.partition("=")
        try:
            value = ast.literal_eval(value)
        except (ValueError, SyntaxError):
            pass
        conf[key] = value
    return conf


def load_config_file(json_file: str, merge_conf: Config = None) -> Config:
    """Load a JSON config file, merging into ``merge_conf`` if given."""
    with open(json_file) as f:
        data = json.load(f)
    conf = merge_conf if merge_conf is not None else Config()
    for k, v in data.items():
        conf[k] = v
    return conf


def save_config(conf: Union[Config, Dict[str, Any]], json_file: str):
    data = conf.data if isinstance(conf, Object) else dict(conf)
    with open(json_file, "w") as f:
        json.dump(data, f, indent=2, default=str)


def merge_config(conf: Config, merge: Union[Dict[str, Any], Config]) -> Config:
    merge_data = merge.data if isinstance(merge, Object) else merge
    for k, v in merge_data.items():
        conf[k] = v
    return conf
