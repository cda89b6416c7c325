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
* ``reduction="reduce_scatter"`` splits each bucket into a
  reduce-scatter + all-gather pair (half the per-link bytes of a ring
  all-reduce on xGMI for large buckets); falls back to all-reduce on
  backends without reduce_scatter support (gloo);
* ``reduction="one_shot"`` all-gathers every rank's bucket and sums
  locally — ONE xGMI hop instead of the ring's 2(N-1)/N round trips
  (every peer is one hop away on this topology, so for
  the reference's SMALL models — CartPole MLPs are a few KB — latency
  dominates and the single-shot direct reduce wins; for large buckets
  it moves N× the bytes, so keep all_reduce/reduce_scatter there);
* no autograd graph rewriting, no "find_unused_parameters" machinery.

Integration contract: after ``backward()`` call :meth:`finalize` (or
let :func:`install_ddp_finalize` hook it into a framework's pluggable
backward), and zero gradients with :meth:`zero_grad_`. Calling
``optimizer.zero_grad(set_to_none=True)`` anyway is SAFE but slower:
the post-accumulate hook detects the detached grad, copies it back
into the bucket and rebinds the view.
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
        reduction: str = "all_reduce",
    ):
        if reduction not in ("all_reduce", "reduce_scatter",
                             "one_shot"):
            raise ValueError(
                "reduction must be 'all_reduce', 'reduce_scatter' or "
                "'one_shot'."
            )
        self.module = module
        self.group = process_group
        self.average = average
        self.reduction = reduction
        self._works: List = []
        self._hooks = []
        self.world_size = (
            dist.get_world_size(self.group) if dist.is_initialized() else 1
        )

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
            padded = total
            if self.reduction == "reduce_scatter" and self.world_size > 1:
                ws = self.world_size
                padded = (total + ws - 1) // ws * ws
            flat = t.zeros(padded, dtype=params[0].dtype, device=device)
            bucket["flat"] = flat
            bucket["ready"] = 0
            bucket["views"] = []
            bucket["view_of"] = {}
            if padded != total or self.reduction == "reduce_scatter":
                bucket["rs_out"] = t.zeros(
                    padded // max(self.world_size, 1),
                    dtype=params[0].dtype, device=device,
                )
            offset = 0
            for p in bucket["params"]:
                view = flat[offset : offset + p.numel()].view_as(p)
                p.grad = view
                bucket["views"].append(view)
                bucket["view_of"][id(p)] = view
                offset += p.numel()

        self._param_bucket = {}
        for bi, bucket in enumerate(self.buckets):
            for p in bucket["params"]:
                self._param_bucket[id(p)] = bi

        for p in params:
            h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
            self._hooks.append(h)

    # ------------------------------------------------------------------
    def _backend_has_reduce_scatter(self) -> bool:
        try:
            backend = dist.get_backend(self.group)
        except Exception:  # noqa: BLE001 - conservative fallback
            return False
        return str(backend) in ("nccl", "rccl")

    def _on_grad_ready(self, param: t.Tensor):
        bi = self._param_bucket[id(param)]
        bucket = self.buckets[bi]
        # survive optimizer.zero_grad(set_to_none=True): autograd then
        # allocated a FRESH grad tensor — fold it into the bucket and
        # rebind the view so the reduction sees the real gradient
        view = bucket["view_of"][id(param)]
        if param.grad is not view:
            with t.no_grad():
                view.copy_(param.grad)
            param.grad = view
        bucket["ready"] += 1
        if bucket["ready"] == len(bucket["params"]):
            bucket["ready"] = 0
            if self.world_size > 1:
                if (
                    self.reduction == "reduce_scatter"
                    and self._backend_has_reduce_scatter()
                ):
                    work = dist.reduce_scatter_tensor(
                        bucket["rs_out"], bucket["flat"],
                        op=dist.ReduceOp.SUM,
                        group=self.group, async_op=True,
                    )
                    self._works.append(("rs", bi, work))
                elif self.reduction == "one_shot":
                    if "gathered" not in bucket:
                        bucket["gathered"] = [
                            t.empty_like(bucket["flat"])
                            for _ in range(self.world_size)
                        ]
                    work = dist.all_gather(
                        bucket["gathered"], bucket["flat"],
                        group=self.group, async_op=True,
                    )
                    self._works.append(("ag", bi, work))
                else:
                    work = dist.all_reduce(
                        bucket["flat"], op=dist.ReduceOp.SUM,
                        group=self.group, async_op=True,
                    )
                    self._works.append(("ar", bi, work))

    def finalize(self):
        """Wait for all in-flight reductions; call after backward().

        Only buckets that were actually reduced in this backward pass
        are averaged — a partial backward (e.g. a loss touching one of
        two wrapped models) must not rescale stale buckets.
        """
        reduced_flats = []
        gather_buckets = []
        for kind, bi, work in self._works:
            work.wait()
            bucket = self.buckets[bi]
            reduced_flats.append(bucket["flat"])
            if kind == "rs":
                gather_buckets.append(bi)
            elif kind == "ag":
                # one-shot: local sum of the gathered per-rank buckets
                t.sum(t.stack(bucket["gathered"]), dim=0,
                      out=bucket["flat"])
        self._works.clear()
        if gather_buckets:
            gathers = [
                dist.all_gather_into_tensor(
                    self.buckets[bi]["flat"], self.buckets[bi]["rs_out"],
                    group=self.group, async_op=True,
                )
                for bi in gather_buckets
            ]
            for w in gathers:
                w.wait()
        if self.world_size > 1 and self.average and reduced_flats:
            t._foreach_mul_(reduced_flats, 1.0 / self.world_size)

    def zero_grad_(self):
        """Zero the flat buffers (keeps the grad views intact — the
        fast equivalent of optimizer.zero_grad())."""
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
    call ``.finalize()`` (or install :func:`install_ddp_finalize` on
    the owning framework)."""

    def __init__(self, module: nn.Module, process_group=None,
                 bucket_cap_mb: float = 32.0,
                 reduction: str = "all_reduce"):
        super().__init__()
        self.module = module
        self.reducer = GradReducer(
            module, process_group, bucket_cap_mb, reduction=reduction
        )
        # broadcast initial params so all ranks start identical
        if self.reducer.world_size > 1:
            src = (
                dist.get_global_rank(process_group, 0)
                if process_group is not None
                else 0
            )
            with t.no_grad():
                for p in module.parameters():
                    dist.broadcast(p.data, src=src, group=process_group)
                for b in module.buffers():
                    dist.broadcast(b.data, src=src, group=process_group)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def finalize(self):
        self.reducer.finalize()

    def zero_grad_(self):
        self.reducer.zero_grad_()


def install_ddp_finalize(frame):
    """Wire a framework's pluggable backward to its GradReducers.

    Fixes the multi-learner integration contract in one place instead
    of per-algorithm: after this call,

    * every ``frame._backward(loss)`` runs autograd backward and then
      ``finalize()``s each wrapped model's reducer (awaits + averages
      the async all-reduces) BEFORE the algorithm calls
      ``optimizer.step()``;
    * each optimizer whose parameters are fully covered by one wrapped
      model zeroes the flat bucket buffers instead of detaching the
      grad views on ``zero_grad(set_to_none=True)``.

    Reference analog: torch DDP does this inside its own backward
    hooks (reference machin/frame/algorithms/apex.py:213-221 relies on
    torch DDP); our reducer keeps it explicit.
    """
    wrapped = []
    for v in frame.__dict__.values():
        if isinstance(v, DistributedDataParallel) and v not in wrapped:
            wrapped.append(v)
    if not wrapped:
        return frame
    reducers = [m.reducer for m in wrapped]

    orig_backward = frame.backward_function

    def ddp_backward(*args, **kwargs):
        orig_backward(*args, **kwargs)
        for r in reducers:
            r.finalize()

    frame.set_backward_function(ddp_backward)

    # optimizer.zero_grad must not detach bucket views
    covered = {}
    for r in reducers:
        for b in r.buckets:
            for p in b["params"]:
                covered[id(p)] = r
    try:
        optimizers = frame.optimizers
    except (NotImplementedError, AttributeError):
        optimizers = []
    for opt in optimizers or []:
        opt_params = [
            p for g in opt.param_groups for p in g["params"]
            if p.requires_grad
        ]
        opt_reducers = {
            covered[id(p)] for p in opt_params if id(p) in covered
        }
        if opt_reducers and all(id(p) in covered for p in opt_params):
            def _zero_grad(set_to_none=True, _rs=tuple(opt_reducers)):
                for r in _rs:
                    r.zero_grad_()

            opt.zero_grad = _zero_grad
    return frame
