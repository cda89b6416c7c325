"""Logging utilities (parity: reference machin/utils/logging.py).

Plain ``logging`` with an ANSI-color formatter; no third-party colorlog
dependency (not present in the ROCm image).
"""
import logging
import sys

_COLORS = {
    "DEBUG": "\x1b[36m",      # cyan
    "INFO": "\x1b[32m",       # green
    "WARNING": "\x1b[33m",    # yellow
    "ERROR": "\x1b[31m",      # red
    "CRITICAL": "\x1b[1;31m", # bold red
}
_RESET = "\x1b[0m"


class _ColorFormatter(logging.Formatter):
    def format(self, record):
        msg = super().format(record)
        color = _COLORS.get(record.levelname, "")
        if color and sys.stderr.isatty():
            return f"{color}{msg}{_RESET}"
        return msg


class FakeLogger:
    """A logger that swallows everything (used in tests / silent mode)."""

    def debug(self, *_, **__):
        pass

    def info(self, *_, **__):
        pass

    def warning(self, *_, **__):
        pass

    def error(self, *_, **__):
        pass

    def critical(self, *_, **__):
        pass

    def exception(self, *_, **__):
        pass

    def setLevel(self, *_, **__):
        pass


fake_logger = FakeLogger()


def _make_default_logger() -> logging.Logger:
    logger = logging.getLogger("machin_amd")
    if not logger.handlers:
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(
            _ColorFormatter("[%(asctime)s] <%(levelname)s>:%(name)s:%(message)s")
        )
        logger.addHandler(handler)
        logger.setLevel(logging.INFO)
        logger.propagate = False
    return logger


default_logger = _make_default_logger()
