"""TensorBoard wrapper.

Parity target: reference ``machin/utils/tensor_board.py`` (:9-33).
The ROCm image has no tensorboardX; torch's own
``torch.utils.tensorboard`` is used when its protobuf deps are
available, otherwise a JSON-lines fallback writer that records
scalars/histogram summaries to disk.
"""
import json
import os
import time
from typing import Any


class _JsonBoardWriter:
    """File-based stand-in exposing the SummaryWriter calls the
    framework uses."""

    def __init__(self, log_dir: str = "runs"):
        os.makedirs(log_dir, exist_ok=True)
        self._file = open(
            os.path.join(log_dir, f"events_{int(time.time())}.jsonl"), "a"
        )

    def _write(self, kind: str, tag: str, value: Any, step):
        self._file.write(
            json.dumps(
                {"kind": kind, "tag": tag, "value": value, "step": step,
                 "time": time.time()}
            )
            + "\n"
        )
        self._file.flush()

    def add_scalar(self, tag, value, global_step=None, **__):
        self._write("scalar", tag, float(value), global_step)

    def add_histogram(self, tag, values, global_step=None, **__):
        import torch as t

        if t.is_tensor(values):
            values = values.detach().float().cpu()
            summary = {
                "mean": float(values.mean()),
                "std": float(values.std()) if values.numel() > 1 else 0.0,
                "min": float(values.min()),
                "max": float(values.max()),
                "numel": values.numel(),
            }
        else:
            summary = {"repr": repr(values)}
        self._write("histogram", tag, summary, global_step)

    def add_text(self, tag, text, global_step=None, **__):
        self._write("text", tag, str(text), global_step)

    def flush(self):
        self._file.flush()

    def close(self):
        self._file.close()


def _make_writer(*args, **kwargs):
    try:
        from torch.utils.tensorboard import SummaryWriter

        return SummaryWriter(*args, **kwargs)
    except Exception:  # noqa: BLE001 - missing protobuf etc.
        return _JsonBoardWriter(*args, **kwargs)


class TensorBoard:
    """Lazily-created global summary writer."""

    def __init__(self):
        self.writer = None

    def init(self, *args, **kwargs):
        if self.writer is None:
            self.writer = _make_writer(*args, **kwargs)
        return self.writer

    def is_inited(self) -> bool:
        return self.writer is not None

    def __getattr__(self, item):
        if self.writer is None:
            raise RuntimeError("TensorBoard not initialized; call init().")
        return getattr(self.writer, item)


default_board = TensorBoard()
