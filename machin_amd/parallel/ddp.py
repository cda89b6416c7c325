"""Bucketed gradient all-reduce over RCCL/xGMI — the framework's
DDP-equivalent.

Replaces the reference's use of ``torch.nn.parallel.
DistributedDataParallel`` for learner groups (reference
machin/frame/algorithms/apex.py:219, impala.py:474-477) with an
explicit reducer designed for one node of 8×MI355X:

* gradients live in flat per-bucket buffers; ``param.grad`` is a VIEW
  into its bucket, so reduction needs no gather/scatter copies;
* buckets are all-reduced asynchronously as soon as their last grad
  arrives (post-accumulate-grad hooks), overlapping communication with
  the rest of backward;
* bucket size defaults to 32 MiB: xGMI is 7 point-to-point links at
  ~153 GB/s, ring all-reduce is per-link bound, so fewer, larger
  collectives beat DDP's default small buckets for these model sizes;
* no autograd graph rewriting, no "find_unused_parameters" machinery —
  call :meth:`finalize` after backward, and use :meth:`zero_grad_`.
"""
from typing import List, Optional

import torch as t
import torch.distributed as dist
import torch.nn as nn


class GradReducer:
    """Flat-bucket gradient all-reducer for one ``nn.Module``."""

    def __init__(
        self,
        module: nn.Module,
        process_group: Optional[dist.ProcessGroup] = None,
        bucket_cap_mb: float = 32.0,
        average: bool = True,
    ):
        self.module = module
        self.group = process_group
        self.average = average
        self._works: List = []
        self._hooks = []

        params = [p for p in module.parameters() if p.requires_grad]
        if not params:
            raise ValueError("Module has no trainable parameters.")
        device = params[0].device
        # allocate in reverse parameter order: backward produces grads
        # roughly output->input, so reverse order fills bucket 0 first
        cap = int(bucket_cap_mb * 1024 * 1024)
        self.buckets: List[dict] = []
        current, size = [], 0
        for p in reversed(params):
            nbytes = p.numel() * p.element_size()
            if current and size + nbytes > cap:
                self.buckets.append({"params": current})
                current, size = [], 0
            current.append(p)
            size += nbytes
        if current:
            self.buckets.append({"params": current})

        for bucket in self.buckets:
            total = sum(p.numel() for p in bucket["params"])
            flat = t.zeros(total, dtype=params[0].dtype, device=device)
            bucket["flat"] = flat
            bucket["ready"] = 0
            offset = 0
            for p in bucket["params"]:
                view = flat[offset : offset + p.numel()].view_as(p)
                p.grad = view
                bucket["views"] = bucket.get("views", []) + [view]
                offset += p.numel()

        self._param_bucket = {}
        for bi, bucket in enumerate(self.buckets):
            for p in bucket["params"]:
                self._param_bucket[id(p)] = bi

        for p in params:
            h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
            self._hooks.append(h)

        self.world_size = (
            dist.get_world_size(self.group) if dist.is_initialized() else 1
        )

    # ------------------------------------------------------------------
    def _on_grad_ready(self, param: t.Tensor):
        bi = self._param_bucket[id(param)]
        bucket = self.buckets[bi]
        bucket["ready"] += 1
        if bucket["ready"] == len(bucket["params"]):
            bucket["ready"] = 0
            if self.world_size > 1:
                work = dist.all_reduce(
                    bucket["flat"], op=dist.ReduceOp.SUM,
                    group=self.group, async_op=True,
                )
                self._works.append(work)

    def finalize(self):
        """Wait for all in-flight reductions; call after backward()."""
        for work in self._works:
            work.wait()
        self._works.clear()
        if self.world_size > 1 and self.average:
            t._foreach_mul_(
                [b["flat"] for b in self.buckets], 1.0 / self.world_size
            )

    def zero_grad_(self):
        """Zero the flat buffers (keeps the grad views intact — do NOT
        use optimizer.zero_grad(set_to_none=True))."""
        t._foreach_zero_([b["flat"] for b in self.buckets])

    def rebind_grads(self):
        """Re-attach grad views (if something detached them)."""
        for bucket in self.buckets:
            for p, v in zip(bucket["params"], bucket["views"]):
                p.grad = v

    def detach(self):
        for h in self._hooks:
            h.remove()
        self._hooks.clear()


class DistributedDataParallel(nn.Module):
    """Thin module wrapper pairing a model with a :class:`GradReducer`.

    API-compatible with the reference's learner wrapping: ``forward``
    delegates, ``.module`` exposes the inner net. After ``backward()``
    call ``.finalize()`` (or use :meth:`sync_context`)."""

    def __init__(self, module: nn.Module, process_group=None,
                 bucket_cap_mb: float = 32.0):
        super().__init__()
        self.module = module
        self.reducer = GradReducer(module, process_group, bucket_cap_mb)
        # broadcast initial params so all ranks start identical
        if self.reducer.world_size > 1:
            with t.no_grad():
                for p in module.parameters():
                    dist.broadcast(p.data, src=0, group=process_group)
                for b in module.buffers():
                    dist.broadcast(b.data, src=0, group=process_group)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def finalize(self):
        self.reducer.finalize()

    def zero_grad_(self):
        self.reducer.zero_grad_()
