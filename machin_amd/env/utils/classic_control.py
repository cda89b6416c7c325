"""Env display utilities.

Parity target: reference ``machin/env/utils/openai_gym.py`` (:1):
``disable_view_window`` — stop gym from opening a render window. The
built-in classic-control envs never open windows, so this is a no-op
kept for API compatibility with scripts written against the
reference.
"""


def disable_view_window() -> None:
    """No-op: machin_amd's built-in environments are headless."""
