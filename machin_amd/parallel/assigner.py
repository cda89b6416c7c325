"""Model-to-device placement optimizer.

Parity target: reference ``machin/parallel/assigner.py``:
``ModelSizeEstimator`` (:10) and ``ModelAssigner`` (:86-372) — a
gradient-descent optimizer over an [n_models x n_devices] placement
probability matrix with a cost combining connection x distance,
capacity overflow, complexity mismatch and entropy.

MI355X note: the device-distance model is re-targeted to one node of
8 GPUs on xGMI: every GPU pair is ONE hop (7 point-to-point links per
GPU), so inter-GPU distance is uniform and small, CPU<->GPU transfers
cost more than GPU<->GPU, and same-device placement costs zero.
"""
from typing import Dict, List, Tuple

import torch as t
import torch.nn as nn


class ModelSizeEstimator:
    """Parameter + buffer size of a model in MiB."""

    def __init__(self, model: nn.Module, size_multiplier: int = 2):
        self.model = model
        self.size_multiplier = size_multiplier

    def get_parameter_sizes(self) -> float:
        return sum(
            p.numel() * p.element_size() for p in self.model.parameters()
        ) / 1024 ** 2

    def get_buffer_sizes(self) -> float:
        return sum(
            b.numel() * b.element_size() for b in self.model.buffers()
        ) / 1024 ** 2

    def estimate_size(self) -> float:
        return (
            self.get_parameter_sizes() + self.get_buffer_sizes()
        ) * self.size_multiplier


def _device_memory_mib(device: t.device) -> float:
    if device.type == "cuda" and t.cuda.is_available():
        free, total = t.cuda.mem_get_info(device)
        return free / 1024 ** 2
    try:
        import psutil

        return psutil.virtual_memory().available / 1024 ** 2
    except ImportError:
        return 16 * 1024.0


def _device_distance(d1: t.device, d2: t.device) -> float:
    """xGMI topology: GPU<->GPU is one hop (0.1), CPU<->GPU crosses
    PCIe/host DRAM (1.0), same device is free."""
    if d1 == d2:
        return 0.0
    if d1.type == "cuda" and d2.type == "cuda":
        return 0.1
    return 1.0


class ModelAssigner:
    """Assign N models to M devices minimizing a soft placement cost."""

    def __init__(
        self,
        models: List[nn.Module],
        model_connection: Dict[Tuple[int, int], int],
        devices: List[t.device] = None,
        model_size_multiplier: int = 2,
        max_mem_ratio: float = 0.5,
        cpu_weight: float = 0.0,
        connection_weight: float = 2.0,
        size_match_weight: float = 1e-2,
        complexity_match_weight: float = 1.0,
        entropy_weight: float = 1.0,
        iterations: int = 500,
        update_rate: float = 0.01,
        gpu_gpu_distance: float = 0.1,
        cpu_gpu_distance: float = 1.0,
        move_models: bool = True,
    ):
        if devices is None:
            devices = [t.device("cpu")] + [
                t.device(f"cuda:{i}") for i in range(t.cuda.device_count())
            ]
        devices = [t.device(d) for d in devices]
        n_models, n_devices = len(models), len(devices)
        self.devices = devices

        sizes = t.tensor(
            [
                ModelSizeEstimator(m, model_size_multiplier).estimate_size()
                for m in models
            ]
        )
        capacity = t.tensor(
            [_device_memory_mib(d) * max_mem_ratio for d in devices]
        )
        dist = t.zeros(n_devices, n_devices)
        for i, d1 in enumerate(devices):
            for j, d2 in enumerate(devices):
                if d1 == d2:
                    dist[i, j] = 0.0
                elif d1.type == "cuda" and d2.type == "cuda":
                    dist[i, j] = gpu_gpu_distance
                else:
                    dist[i, j] = cpu_gpu_distance
        # per-device relative compute complexity: GPUs equal, CPU weak
        complexity = t.tensor(
            [cpu_weight if d.type == "cpu" else 1.0 for d in devices]
        )
        complexity = complexity / complexity.sum().clamp_min(1e-6)

        # optimize placement probabilities with Adam
        logits = t.randn(n_models, n_devices, requires_grad=True)
        optim = t.optim.Adam([logits], lr=update_rate)
        for _ in range(iterations):
            prob = t.softmax(logits, dim=1)
            # expected pairwise transfer cost
            conn_cost = t.zeros(())
            for (a, b), strength in model_connection.items():
                pa, pb = prob[a], prob[b]
                conn_cost = conn_cost + strength * (
                    pa.unsqueeze(1) * pb.unsqueeze(0) * dist
                ).sum()
            # capacity overflow
            exp_load = prob.t() @ sizes
            over_cost = t.relu(exp_load - capacity).sum()
            # complexity match: spread model mass like device compute
            mass = prob.sum(dim=0) / n_models
            complexity_cost = (mass - complexity).abs().sum()
            # low entropy -> decisive placement
            entropy_cost = -(prob * (prob + 1e-8).log()).sum(dim=1).mean()
            loss = (
                connection_weight * conn_cost
                + size_match_weight * over_cost
                + complexity_match_weight * complexity_cost
                + entropy_weight * entropy_cost
            )
            optim.zero_grad()
            loss.backward()
            optim.step()

        with t.no_grad():
            self._assignment = t.softmax(logits, dim=1).argmax(dim=1)
        if move_models:
            for m, di in zip(models, self._assignment.tolist()):
                m.to(devices[di])

    @property
    def assignment(self) -> List[t.device]:
        return [self.devices[i] for i in self._assignment.tolist()]
