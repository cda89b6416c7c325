from .helper_classes import Counter, Object, Switch, Timer, Trigger
from .logging import default_logger, fake_logger

__all__ = [
    "Counter",
    "Object",
    "Switch",
    "Timer",
    "Trigger",
    "default_logger",
    "fake_logger",
]

from .conf import Config, load_config_cmd, load_config_file, save_config
from .learning_rate import gen_learning_rate_func
from .prepare import prep_create_dirs, prep_load_model, prep_load_state_dict
from .save_env import SaveEnv

__all__ += [
    "Config",
    "load_config_cmd",
    "load_config_file",
    "save_config",
    "SaveEnv",
    "prep_create_dirs",
    "prep_load_model",
    "prep_load_state_dict",
    "gen_learning_rate_func",
]
