"""Local media/metrics logger.

Parity target: reference ``machin/auto/pl_logger.py``
(``LocalMediaLogger`` :12): collects artifact files (rendered episode
videos/images) into a local directory and scalar metrics into a
JSON-lines file.
"""
import json
import os
import shutil
import time


class LocalMediaLogger:
    def __init__(self, image_dir: str, artifact_dir: str):
        self.image_dir = image_dir
        self.artifact_dir = artifact_dir
        os.makedirs(image_dir, exist_ok=True)
        os.makedirs(artifact_dir, exist_ok=True)
        self._metrics = open(
            os.path.join(artifact_dir, "metrics.jsonl"), "a"
        )

    def log_metrics(self, metrics: dict, step=None):
        self._metrics.write(
            json.dumps({"step": step, "time": time.time(), **metrics}) + "\n"
        )
        self._metrics.flush()

    def log_artifact(self, local_path: str, artifact_path: str = None):
        if not os.path.exists(local_path):
            return
        dest = os.path.join(
            self.artifact_dir, artifact_path or os.path.basename(local_path)
        )
        if os.path.abspath(local_path) != os.path.abspath(dest):
            shutil.copy(local_path, dest)

    def finalize(self):
        self._metrics.close()
