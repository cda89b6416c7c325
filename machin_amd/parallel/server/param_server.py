"""Parameter servers: whole-model push/pull and gradient reduction.

Parity target: reference ``machin/parallel/server/param_server.py``:

* ``PushPullModelServer(+Impl)`` (:21-141) — versioned whole-state-dict
  sync on an ordered KV server; ``push`` bumps ``model.pp_version`` and
  retries by pulling on version conflict; ``pull`` loads only if newer.
* ``PushPullGradServer(+Impl)`` (:148-493) — two-level gradient
  reduction: clients push grad dicts to a random secondary reducer;
  each secondary batches ``reduce_batch_size`` dicts, reduces
  per-parameter, forwards to the primary; the primary applies the
  reduced grads to its managed model, steps the optimizer and
  publishes fresh parameters.

MI355X note: these servers carry CPU state dicts over the TCP control
plane — correct for host-side actor farms (A3C/ARS). GPU learner
groups should NOT route gradients here; they use
machin_amd.parallel.ddp.GradReducer (bucketed RCCL all-reduce over
xGMI), as APEX/IMPALA do.
"""
import enum
import random
import threading
import time
from typing import Callable, Dict

import torch as t
import torch.nn as nn

from ..distributed.world import RpcGroup
from .ordered_server import (
    OrderedServerBase,
    OrderedServerSimple,
    OrderedServerSimpleImpl,
)


class PushPullModelServer:
    """Accessor for whole-model push/pull."""

    def __init__(self, model_name: str, o_server: OrderedServerBase):
        self.model_name = model_name
        self.o_server = o_server

    def push(self, model: nn.Module, pull_on_fail: bool = True) -> bool:
        if not hasattr(model, "pp_version"):
            model.pp_version = 0
        state = {
            k: v.detach().cpu().clone()
            for k, v in model.state_dict().items()
        }
        version = model.pp_version + 1
        ok = self.o_server.push(
            self.model_name, state, version=version,
            prev_version=model.pp_version,
        )
        if ok:
            model.pp_version = version
        elif pull_on_fail:
            self.pull(model)
        return ok

    def pull(self, model: nn.Module) -> bool:
        result = self.o_server.pull(self.model_name)
        if result is None:
            return False
        state, version = result
        if getattr(model, "pp_version", -1) >= version:
            return True
        own = model.state_dict()
        model.load_state_dict(
            {
                k: v.to(own[k].device) if t.is_tensor(v) else v
                for k, v in state.items()
            }
        )
        model.pp_version = version
        return True


class PushPullModelServerImpl:
    """Construct on ONE member; creates the backing ordered server."""

    def __init__(self, server_name: str, group: RpcGroup,
                 model_name: str = "model"):
        self.server_name = server_name
        self.group = group
        self._o_server_impl = OrderedServerSimpleImpl(
            server_name + "_o_server", group
        )
        accessor = PushPullModelServer(
            model_name, OrderedServerSimple(server_name + "_o_server", group)
        )
        group.pair(server_name, accessor)


class ReduceType(enum.Enum):
    REDUCE_PRIMARY = 0
    REDUCE_SECONDARY = 1


class PushPullGradServer:
    """Accessor: push gradients, pull fresh parameters."""

    def __init__(self, server_name: str, group: RpcGroup,
                 model_name: str, secondary_reducers, o_server):
        self.server_name = server_name
        self.group = group
        self.model_name = model_name
        self.secondary_reducers = list(secondary_reducers)
        self.o_server = o_server

    def push(self, model: nn.Module):
        """Send this model's gradients to a random secondary reducer,
        then refresh the model's parameters."""
        grads = {}
        for name, p in model.named_parameters():
            if p.grad is not None:
                grads[name] = p.grad.detach().cpu().clone()
        target = random.choice(self.secondary_reducers)
        self.group.registered_sync(
            self.server_name + f"/{target}/_push_service", args=(grads,)
        )
        self.pull(model)

    def pull(self, model: nn.Module) -> bool:
        result = self.o_server.pull(self.model_name)
        if result is None:
            return False
        state, version = result
        if getattr(model, "pp_version", -1) >= version:
            return True
        own = model.state_dict()
        model.load_state_dict(
            {
                k: v.to(own[k].device) if t.is_tensor(v) else v
                for k, v in state.items()
            }
        )
        model.pp_version = version
        return True


class PushPullGradServerImpl:
    """Construct on EVERY member of the reduce group.

    The first member is the primary: it holds the managed model and
    optimizer and publishes parameters; every member (including the
    primary) also runs a secondary reducer that batches client
    gradient dicts before forwarding one reduced dict upstream.
    """

    def __init__(
        self,
        server_name: str,
        group: RpcGroup,
        model_name: str = "model",
        reduce_method: str = "sum",
        reduce_device="cpu",
        reduce_batch_size: int = 4,
        max_queue_size: int = 64,
        reducer_members=None,
        reduce_timeout: float = 0.2,
    ):
        if reduce_method not in ("sum", "mean"):
            raise ValueError("reduce_method must be 'sum' or 'mean'.")
        self.server_name = server_name
        self.group = group
        self.model_name = model_name
        self.reduce_method = reduce_method
        self.reduce_batch_size = reduce_batch_size
        self.max_queue_size = max_queue_size
        self.reduce_timeout = reduce_timeout
        self._last_push = time.monotonic()
        self.members = list(reducer_members or group.get_group_members())
        self.primary = self.members[0]
        self.me = group.get_cur_name()
        self.is_primary = self.me == self.primary

        self.model: nn.Module = None
        self.optimizer = None
        self._model_lock = threading.Lock()

        self._queue = []
        self._queue_lock = threading.Lock()
        self._stop = threading.Event()

        # every member runs a secondary reducer service
        group.register(
            server_name + f"/{self.me}/_push_service", self._push_service
        )
        if self.is_primary:
            group.register(
                server_name + "/_master_push_service",
                self._master_push_service,
            )
            self._o_server_impl = OrderedServerSimpleImpl(
                server_name + "_o_server", group
            )
            o_server = OrderedServerSimple(server_name + "_o_server", group)
            accessor = PushPullGradServer(
                server_name, group, model_name, self.members, o_server
            )
            group.pair(server_name, accessor)

        self._reduce_thread = threading.Thread(
            target=self._reduce_loop, daemon=True
        )
        self._reduce_thread.start()

    # ------------------------------------------------------------------
    def manage_model(self, model: nn.Module, optimizer):
        """Primary only: attach the model updated by reduced grads."""
        if not self.is_primary:
            raise RuntimeError("Only the primary reducer manages a model.")
        with self._model_lock:
            self.model = model
            self.optimizer = optimizer
        self._publish()

    def watch(self):
        pass  # reduce thread is a daemon; errors surface via logging

    def stop(self):
        self._stop.set()

    # ------------------------------------------------------------------
    def _push_service(self, grads: Dict[str, t.Tensor]):
        with self._queue_lock:
            if len(self._queue) >= self.max_queue_size:
                self._queue.pop(0)
            self._queue.append(grads)
            self._last_push = time.monotonic()
        return True

    def _master_push_service(self, grads: Dict[str, t.Tensor], count: int):
        self._apply(grads, count)
        return True

    def _reduce_loop(self):
        while not self._stop.is_set():
            batch = None
            with self._queue_lock:
                if len(self._queue) >= self.reduce_batch_size:
                    batch = self._queue[: self.reduce_batch_size]
                    del self._queue[: self.reduce_batch_size]
                elif (
                    self._queue
                    and time.monotonic() - self._last_push
                    > self.reduce_timeout
                ):
                    # flush a partial batch so slow/asymmetric pushers
                    # cannot stall the whole reduction pipeline
                    batch = self._queue[:]
                    self._queue.clear()
            if batch is None:
                time.sleep(1e-3)
                continue
            reduced = self._reduce_batch(batch)
            if self.is_primary:
                self._apply(reduced, len(batch))
            else:
                self.group.registered_sync(
                    self.server_name + "/_master_push_service",
                    args=(reduced, len(batch)),
                )

    def _reduce_batch(self, batch):
        """Per-parameter stacked reduce of a list of grad dicts."""
        keys = batch[0].keys()
        out = {}
        for k in keys:
            stacked = t.stack([g[k] for g in batch if k in g])
            out[k] = stacked.sum(dim=0)
            if self.reduce_method == "mean":
                out[k] = out[k] / stacked.shape[0]
        return out

    def _apply(self, grads: Dict[str, t.Tensor], count: int):
        # grads arriving here are already reduced by _reduce_batch
        # (which divides by the batch count when reduce_method is
        # "mean") — scaling again here would shrink applied gradients
        # by ~count×. ``count`` is kept for the
        # service signature only.
        with self._model_lock:
            if self.model is None:
                return
            params = dict(self.model.named_parameters())
            for k, g in grads.items():
                if k in params:
                    params[k].grad = g.to(params[k].device)
            self.optimizer.step()
            self.optimizer.zero_grad(set_to_none=False)
            self._publish()

    def _publish(self):
        if self.model is None:
            return
        if not hasattr(self.model, "pp_version"):
            self.model.pp_version = 0
        state = {
            k: v.detach().cpu().clone()
            for k, v in self.model.state_dict().items()
        }
        version = self.model.pp_version + 1
        ok = self._o_server_impl._push_service(
            self.model_name, state, version, self.model.pp_version
        )
        if ok:
            self.model.pp_version = version
