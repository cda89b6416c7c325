"""Trial directory scaffolding.

Parity target: reference ``machin/utils/save_env.py`` (:12-52):
creates ``root/<timestamp>/{config,log/images,log/train_log,model}``
per trial, with ``restart_from_trial`` reusing an existing trial dir.
"""
import os
import time
from typing import List


class SaveEnv:
    def __init__(self, env_root: str, restart_from_trial: str = None,
                 time_format: str = "%Y_%m_%d_%H_%M_%S"):
        self.env_root = env_root
        self._time_format = time_format
        if restart_from_trial is not None:
            self.env_create_time = restart_from_trial
        else:
            self.env_create_time = time.strftime(time_format)
        self._trial_root = os.path.join(env_root, self.env_create_time)
        self.create_dirs(
            ["config", os.path.join("log", "images"),
             os.path.join("log", "train_log"), "model"]
        )

    def create_dirs(self, dirs: List[str]):
        for d in dirs:
            os.makedirs(os.path.join(self._trial_root, d), exist_ok=True)

    def get_trial_root(self) -> str:
        return self._trial_root

    def get_trial_config_dir(self) -> str:
        return os.path.join(self._trial_root, "config")

    def get_trial_model_dir(self) -> str:
        return os.path.join(self._trial_root, "model")

    def get_trial_image_dir(self) -> str:
        return os.path.join(self._trial_root, "log", "images")

    def get_trial_train_log_dir(self) -> str:
        return os.path.join(self._trial_root, "log", "train_log")

    def clear_trial_config_dir(self):
        self._clear(self.get_trial_config_dir())

    def clear_trial_model_dir(self):
        self._clear(self.get_trial_model_dir())

    def clear_trial_image_dir(self):
        self._clear(self.get_trial_image_dir())

    def clear_trial_train_log_dir(self):
        self._clear(self.get_trial_train_log_dir())

    def remove_trials_older_than(
        self, diff_day=0, diff_hour=1, diff_minute=0, diff_second=0
    ):
        """Delete trial dirs whose timestamp is older than the given
        age."""
        import shutil

        threshold = time.time() - (
            ((diff_day * 24 + diff_hour) * 60 + diff_minute) * 60
            + diff_second
        )
        if not os.path.isdir(self.env_root):
            return
        for name in os.listdir(self.env_root):
            path = os.path.join(self.env_root, name)
            if not os.path.isdir(path) or name == self.env_create_time:
                continue
            try:
                created = time.mktime(
                    time.strptime(name, self._time_format)
                )
            except ValueError:
                continue
            if created < threshold:
                shutil.rmtree(path, ignore_errors=True)

    @staticmethod
    def _clear(directory: str):
        import shutil

        for name in os.listdir(directory):
            path = os.path.join(directory, name)
            if os.path.isdir(path):
                shutil.rmtree(path, ignore_errors=True)
            else:
                os.remove(path)
