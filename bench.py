"""Flagship benchmark: IMPALA Atari-CNN end-to-end throughput on
MI355X — BOTH halves of the BASELINE.json headline metric:

* ``learner samples/sec`` — samples consumed by gradient updates
  (u8 dequant, bf16 forward, V-trace gfx950 kernel, losses, backward,
  bucketed RCCL all-reduce over xGMI for N>1, clip, optimizer step);
* ``env-steps/sec`` — fresh environment steps ingested while training
  runs: a host actor farm writes rollout segments into a
  shared-memory RolloutRing; a background thread drains them through
  pinned staging (hipHostRegister direct DMA when available) onto a
  side stream into an HBM-resident segment pool the learner trains
  from.

Default mode is the full pipeline; ``--learner-only`` measures the
bare learner loop from GPU-resident pools (round-1 configuration).

Usage:
    python bench.py [--gpus N] [--steps K] [--warmup W]
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Scaling is WEAK: per-GPU work (unroll x env_batch) is fixed as N grows.
"""
import argparse
import json
import os
import sys
import threading
import time

import torch as t
import torch.distributed as dist
import torch.nn as nn


def _actor_loop(ring, actor_id, stop_flag, steps_counter, unroll,
                frames, actions):
    """Synthetic Atari actor process: fills rollout segments in shared
    memory. A real actor would run ALE + a policy net (no ROMs without
    network access); a pre-generated segment library is cycled with
    cheap mutation so the measured cost is the TRANSPORT pipeline
    (shared-memory writes, slot bookkeeping), not RNG."""
    t.set_num_threads(1)
    try:
        os.nice(5)  # the learner's host threads have priority
    except OSError:
        pass
    gen = t.Generator().manual_seed(actor_id)
    library = [
        {
            "frames": t.randint(0, 256, (unroll, frames, 84, 84),
                                dtype=t.uint8, generator=gen),
            "actions": t.randint(0, actions, (unroll,), generator=gen),
            "behavior_logp": -t.rand(unroll, generator=gen) * 2.0,
            "rewards": t.rand(unroll, generator=gen),
            "terminals": t.zeros(unroll),
        }
        for _ in range(8)
    ]
    i = 0
    while not stop_flag[0]:
        try:
            slot_id = ring.acquire(timeout=1.0)
        except Exception:  # noqa: BLE001 - queue.Empty on shutdown
            continue
        src = library[i % len(library)]
        i += 1
        slot = ring.slot(slot_id)
        slot["frames"].copy_(src["frames"])
        slot["frames"][0, 0, 0, 0] = i % 256  # per-segment variation
        slot["actions"].copy_(src["actions"])
        slot["behavior_logp"].copy_(src["behavior_logp"])
        slot["rewards"].copy_(src["rewards"])
        slot["terminals"].copy_(src["terminals"])
        ring.commit(slot_id)
        with steps_counter.get_lock():
            steps_counter.value += unroll


class RingPipeline:
    """Actor farm -> shared-memory ring -> pinned staging -> HBM pool
    (the BASELINE "async rollout staging" path), feeding the learner
    with t-major batches gathered on device."""

    def __init__(self, device, unroll, env_batch, actions, frames=4,
                 actors=6, slots=256, pool_segments=4096):
        import torch.multiprocessing as mp

        from machin_amd.parallel.rollout_ring import RolloutRing

        self.device = device
        self.unroll = unroll
        self.env_batch = env_batch
        ctx = mp.get_context("spawn")
        self.spec = {
            "frames": ((unroll, frames, 84, 84), t.uint8),
            "actions": ((unroll,), t.long),
            "behavior_logp": ((unroll,), t.float32),
            "rewards": ((unroll,), t.float32),
            "terminals": ((unroll,), t.float32),
        }
        self.ring = RolloutRing(slots=slots, spec=self.spec, ctx=ctx)
        self.stop_flag = t.zeros(1, dtype=t.uint8).share_memory_()
        self.steps_counter = ctx.Value("q", 0)
        self.actors = [
            ctx.Process(
                target=_actor_loop,
                args=(self.ring, a, self.stop_flag, self.steps_counter,
                      unroll, frames, actions),
                daemon=True,
            )
            for a in range(actors)
        ]
        M = pool_segments
        self.M = M
        # frames pool is stored NHWC per step so a training gather is
        # a ZERO-COPY view (the [T, B] layout forced a 1.2 GB
        # transpose per step on the critical path); the permute to
        # NHWC happens during ingest on the side stream instead
        self.pool = {
            k: t.zeros((M, *shape), dtype=dtype, device=device)
            for k, (shape, dtype) in self.spec.items()
            if k != "frames"
        }
        self.pool["frames"] = t.zeros(
            (M, unroll, 84, 84, frames), dtype=t.uint8, device=device
        )
        self._frames_stage = t.empty(
            (64, unroll, frames, 84, 84), dtype=t.uint8, device=device
        )
        self.pinned = self.ring.make_pinned_staging(64)
        self.registered = False
        self.side = t.cuda.Stream()
        self.filled = 0
        self.write_pos = 0
        self._stop_ingest = False
        self._pause_ingest = False
        self._ingester = None
        self.gather_stream = t.cuda.Stream()
        self._next = None
        self._next_ev = None

    def start(self):
        for p in self.actors:
            p.start()
        self.registered = self.ring.host_register()
        # pre-fill the pool so the first batches sample real segments;
        # bounded so a stalled actor farm degrades to a partial pool
        # (or a clear error) instead of hanging the driver's run
        deadline = time.monotonic() + 120.0
        target = min(self.env_batch, self.M)
        while self.filled < target:
            self.ingest_once(timeout=2.0)
            if time.monotonic() > deadline:
                if self.filled == 0:
                    raise RuntimeError(
                        "rollout actors produced nothing within 120s"
                    )
                print(
                    f"[bench] pool prefill timed out at "
                    f"{self.filled}/{target} segments; continuing",
                    file=sys.stderr,
                )
                break
        self._ingester = threading.Thread(
            target=self._ingest_loop, daemon=True
        )
        self._ingester.start()

    def ingest_once(self, timeout=0.05):
        idx = self.ring.drain(max_slots=64, timeout=timeout)
        if not idx:
            return 0
        wp, M, n = self.write_pos, self.M, len(idx)
        positions = [(wp + i) % M for i in range(n)]
        with t.cuda.stream(self.side):
            if self.registered:
                small = {
                    k: v for k, v in self.pool.items() if k != "frames"
                }
                self.ring.upload_slots(idx, small, positions)
                # frames: direct-DMA into TCHW staging, then one
                # permuted device copy into the NHWC pool rows
                for j, (slot_id, pos) in enumerate(zip(idx, positions)):
                    self._frames_stage[j].copy_(
                        self.ring.data["frames"][slot_id],
                        non_blocking=True,
                    )
                stage = self._frames_stage[:n]
                pos_t = t.tensor(positions, device=self.device)
                self.pool["frames"].index_copy_(
                    0, pos_t, stage.permute(0, 1, 3, 4, 2).contiguous()
                )
            else:
                batch = self.ring.gather(
                    idx, self.device, pinned=self.pinned
                )
                for k, v in batch.items():
                    if k == "frames":
                        v = v.permute(0, 1, 3, 4, 2).contiguous()
                    end = wp + n
                    if end <= M:
                        self.pool[k][wp:end] = v
                    else:
                        split = M - wp
                        self.pool[k][wp:] = v[:split]
                        self.pool[k][: end % M] = v[split:]
        self.side.synchronize()
        self.ring.release(idx)
        self.write_pos = (wp + n) % M
        self.filled = min(self.filled + n, M)
        return n

    def _ingest_loop(self):
        # background staging, overlapped with training (host memcpys
        # release the GIL)
        while not self._stop_ingest:
            if self._pause_ingest:
                time.sleep(0.005)
                continue
            self.ingest_once()

    def sample(self, out=None):
        """Gather a BATCH-MAJOR training batch from the HBM pool. The
        frames pool is NHWC segment-major, so the gathered [B, T, H,
        W, C] block IS the channels-last [B*T, C, H, W] image — one
        index_select, zero transpose copies (the t-major layout paid
        a 1.2 GB strided copy per step). With ``out`` the gather
        writes into preallocated static buffers (hipGraph mode)."""
        B, T = self.env_batch, self.unroll
        seg = t.randint(0, max(self.filled, 1), (B,), device=self.device)
        p = self.pool
        if out is None:
            fr = p["frames"].index_select(0, seg)  # [B, T, H, W, C]
            H, W, C = fr.shape[2:]
            out = {
                "frames": fr.view(B * T, H, W, C).permute(0, 3, 1, 2)
            }
            for k in ("actions", "behavior_logp", "rewards",
                      "terminals"):
                out[k] = p[k].index_select(0, seg)  # [B, T]
            return out
        H, W, C = p["frames"].shape[2:]
        out["frames"].view(B, T, C, H, W).permute(0, 1, 3, 4, 2).copy_(
            p["frames"].index_select(0, seg)
        )
        for k in ("actions", "behavior_logp", "rewards", "terminals"):
            out[k].copy_(p[k].index_select(0, seg))
        return out

    def sample_prefetched(self):
        """Double-buffered sample: the NEXT batch's gather runs on a
        separate stream, overlapped with the current step's compute
        (the serial gather cost ~1 ms/step of the e2e gap vs
        learner-only)."""
        cur = t.cuda.current_stream()
        if self._next is None:
            with t.cuda.stream(self.gather_stream):
                self._next = self.sample()
            self._next_ev = t.cuda.Event()
            self._next_ev.record(self.gather_stream)
        batch, ev = self._next, self._next_ev
        cur.wait_event(ev)
        for v in batch.values():
            # tensors were allocated on gather_stream; tell the caching
            # allocator they are consumed on the compute stream
            v.record_stream(cur)
        with t.cuda.stream(self.gather_stream):
            self._next = self.sample()
        self._next_ev = t.cuda.Event()
        self._next_ev.record(self.gather_stream)
        return batch

    def make_static_batch(self):
        B, T = self.env_batch, self.unroll
        H, W, C = self.pool["frames"].shape[2:]
        return {
            "frames": t.empty(
                (B * T, C, H, W), dtype=t.uint8, device=self.device,
                memory_format=t.channels_last,
            ),
            "actions": t.empty((B, T), dtype=t.long,
                               device=self.device),
            "behavior_logp": t.empty((B, T), device=self.device),
            "rewards": t.empty((B, T), device=self.device),
            "terminals": t.empty((B, T), device=self.device),
        }

    def reset_env_steps(self):
        with self.steps_counter.get_lock():
            self.steps_counter.value = 0

    def env_steps(self) -> int:
        with self.steps_counter.get_lock():
            return self.steps_counter.value

    def stop(self):
        self._stop_ingest = True
        if self._ingester is not None:
            self._ingester.join(timeout=5)
        self.stop_flag[0] = 1
        for p in self.actors:
            p.join(timeout=5)
            if p.is_alive():
                p.terminate()


class ImpalaLearnerBench:
    """One IMPALA learner replica (one GPU)."""

    def __init__(
        self,
        device="cuda:0",
        unroll: int = 20,
        env_batch: int = 2048,
        action_num: int = 6,
        frames: int = 4,
        dtype=t.bfloat16,
        pool_size: int = 4,
        distributed: bool = False,
        lr: float = 6e-4,
        entropy_weight: float = 0.01,
        value_weight: float = 0.5,
        discount: float = 0.99,
        grad_clip: float = 40.0,
        capturable: bool = False,
        fused_stem: bool = True,
        bucket_mb: float = 32.0,
        ddp_reduction: str = "all_reduce",
        pipeline: RingPipeline = None,
    ):
        from machin_amd.model.nets.nature_cnn import (
            ActorCriticCNN,
            FusedActorCriticCNN,
        )
        import machin_amd.ops as ops

        if not ops.available():
            raise RuntimeError(
                "machin_amd HIP extension not built — refusing to benchmark "
                "the eager fallback. Run: python setup.py build_ext --inplace"
            )
        self.ops = ops
        self.device = t.device(device)
        self.unroll = unroll
        self.env_batch = env_batch
        self.dtype = dtype
        self.entropy_weight = entropy_weight
        self.value_weight = value_weight
        self.discount = discount
        self.grad_clip = grad_clip
        self.pipeline = pipeline

        self.fused_stem = fused_stem and dtype == t.bfloat16 and frames == 4
        if self.fused_stem:
            self.model = FusedActorCriticCNN(action_num).to(self.device)
        else:
            self.model = ActorCriticCNN(frames, action_num).to(self.device)
        self.model = self.model.to(memory_format=t.channels_last)
        self.reducer = None
        if distributed:
            from machin_amd.parallel.ddp import GradReducer

            self.reducer = GradReducer(
                self.model, bucket_cap_mb=bucket_mb,
                reduction=ddp_reduction,
            )
            with t.no_grad():
                for p in self.model.parameters():
                    dist.broadcast(p.data, src=0)
        self.optim = t.optim.RMSprop(
            self.model.parameters(), lr=lr, alpha=0.99, eps=0.1,
            capturable=capturable,
        )
        self.capturable = capturable
        self._fused_opt = None

        # synthetic rollout pool for --learner-only mode (uint8 frames
        # like a real Atari actor feed)
        self.pool = []
        if pipeline is None:
            TB = unroll * env_batch
            # per-rank data seeds: ranks must NOT train on identical
            # data, or the all-reduce degenerates to a no-op check
            rank = int(os.environ.get("RANK", "0"))
            g = t.Generator(device="cpu").manual_seed(1234 + rank * 1000)
            for _ in range(pool_size):
                self.pool.append(
                    {
                        "frames": t.randint(
                            0, 256, (TB, frames, 84, 84), dtype=t.uint8,
                            generator=g,
                        ).to(self.device)
                        .to(memory_format=t.channels_last),
                        "actions": t.randint(
                            0, action_num, (env_batch, unroll),
                            generator=g,
                        ).to(self.device),
                        "behavior_logp": (
                            -t.rand(env_batch, unroll, generator=g) * 2.0
                        ).to(self.device),
                        "rewards": t.rand(
                            env_batch, unroll, generator=g
                        ).to(self.device),
                        "terminals": (
                            t.rand(env_batch, unroll, generator=g) > 0.98
                        ).float().to(self.device),
                    }
                )
        self._pool_i = 0
        self._graph = None
        self._static_in = None
        self._static_loss = None

    def step(self):
        """One learner step; returns the detached loss TENSOR (no
        host synchronization on the hot path)."""
        if self.pipeline is not None:
            if self._graph is not None:
                # gather straight into the graph's static buffers,
                # then one replay instead of ~300 launches
                self.pipeline.sample(out=self._static_in)
                self._graph.replay()
                return self._static_loss
            return self._step_body(self.pipeline.sample_prefetched())
        data = self.pool[self._pool_i]
        self._pool_i = (self._pool_i + 1) % len(self.pool)
        if self._graph is not None:
            for k, v in self._static_in.items():
                v.copy_(data[k], non_blocking=True)
            self._graph.replay()
            return self._static_loss
        return self._step_body(data)

    def _step_body(self, data):
        T, B = self.unroll, self.env_batch

        if self.fused_stem:
            # raw u8 frames straight into the fused stem (dequant and
            # the stem conv weight gradient are hand-written kernels)
            logits, values = self.model(data["frames"])
        else:
            if self.dtype == t.bfloat16:
                nhwc = data["frames"].permute(0, 2, 3, 1)
                out = self.ops.dequant_u8(nhwc, 1.0 / 255.0)
                frames = out.view(nhwc.shape).permute(0, 3, 1, 2)
            else:
                frames = data["frames"].to(self.dtype).mul_(1.0 / 255.0)
            with t.autocast(device_type="cuda", dtype=self.dtype):
                logits, values = self.model(frames)
        # everything batch-major [B, T] (flat row = b*T + t): the
        # frames gather is then a zero-copy view of the NHWC pool
        values = values.float().view(B, T)
        # fused policy head: ONE kernel for log_softmax+gather+entropy
        # (and one analytic backward) instead of the eager chain
        tl_flat, ent_flat = self.ops.categorical_policy_head(
            logits.reshape(B * T, -1), data["actions"].reshape(-1)
        )
        taken_logp = tl_flat.view(B, T)

        with t.no_grad():
            bootstrap = values[:, -1].detach().contiguous()
            vs, pg_adv = self.ops.vtrace(
                data["behavior_logp"],
                taken_logp.detach(),
                data["rewards"],
                values.detach(),
                bootstrap,
                data["terminals"],
                self.discount,
                time_major=False,
            )

        pg_loss = -(pg_adv * taken_logp).sum() / B
        value_loss = 0.5 * ((vs - values) ** 2).sum() / B
        entropy = ent_flat.sum() / B
        loss = (
            pg_loss
            + self.value_weight * value_loss
            - self.entropy_weight * entropy
        )

        if self.reducer is not None:
            self.reducer.zero_grad_()
        else:
            self.optim.zero_grad(set_to_none=False)
        loss.backward()
        if self.reducer is not None:
            self.reducer.finalize()
        # fused clip+RMSprop (2 launches, no host sync); the first
        # step runs eagerly to materialize the optimizer state the
        # plan's pointer table needs
        if self._fused_opt is not None and self._fused_opt.matches(
            self.optim
        ):
            self._fused_opt.step()
        else:
            nn.utils.clip_grad_norm_(
                self.model.parameters(), self.grad_clip
            )
            self.optim.step()
            if not self.capturable:
                try:
                    self._fused_opt = self.ops.FusedRMSprop(
                        self.optim, max_norm=self.grad_clip
                    )
                except (ValueError, RuntimeError):
                    self._fused_opt = None
        return loss.detach()

    def capture_graph(self, warmup_steps: int = 3):
        """Capture the whole learner step in a hipGraph: the Nature
        CNN is small, so kernel-launch overhead is a real cost at low
        batch — one graph replay replaces ~300 launches. Single-GPU
        only (RCCL collectives stay outside graphs); works in both
        learner-only and pipeline mode (the pool gather runs eagerly
        into the graph's static input buffers each step)."""
        if self.reducer is not None:
            raise RuntimeError("graph capture is single-GPU only")
        if self.pipeline is not None:
            # concurrent side-stream ingest breaks stream capture
            self.pipeline._pause_ingest = True
            time.sleep(0.1)
            t.cuda.synchronize()
            self._static_in = self.pipeline.make_static_batch()
            self.pipeline.sample(out=self._static_in)
        else:
            data = self.pool[0]
            self._static_in = {
                k: v.clone() for k, v in data.items()
            }
        side = t.cuda.Stream()
        side.wait_stream(t.cuda.current_stream())
        with t.cuda.stream(side):
            for _ in range(warmup_steps):
                self._step_body(self._static_in)
        t.cuda.current_stream().wait_stream(side)
        self._graph = t.cuda.CUDAGraph()
        with t.cuda.graph(self._graph):
            self._static_loss = self._step_body(self._static_in)
        if self.pipeline is not None:
            self.pipeline._pause_ingest = False


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=150)
    parser.add_argument("--warmup", type=int, default=30)
    parser.add_argument("--unroll", type=int, default=20)
    parser.add_argument("--env-batch", type=int, default=2048)
    parser.add_argument("--actions", type=int, default=6)
    parser.add_argument("--actors", type=int, default=0,
                        help="actor processes per rank (0 = auto)")
    parser.add_argument("--slots", type=int, default=256)
    parser.add_argument("--pool-segments", type=int, default=4096)
    parser.add_argument("--learner-only", action="store_true",
                        help="skip the actor farm/ring; measure the "
                             "bare learner loop from resident pools")
    parser.add_argument("--graph", action="store_true",
                        help="capture the learner step in a hipGraph "
                             "(opt-in; measured net-negative here)")
    parser.add_argument("--no-graph", action="store_true",
                        help="disable hipGraph capture")
    parser.add_argument("--bucket-mb", type=float, default=32.0,
                        help="GradReducer bucket size (MiB)")
    parser.add_argument("--ddp-reduction", default="all_reduce",
                        choices=["all_reduce", "reduce_scatter", "one_shot"])
    parser.add_argument("--no-fused-stem", action="store_true",
                        help="fall back to MIOpen for the stem conv")
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world_size > 1
    if distributed:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group("nccl", rank=rank, world_size=world_size)
    if not distributed and args.gpus > 1:
        raise SystemExit(
            "--gpus N>1 requires torchrun (one rank per GPU): "
            "python -m torch.distributed.run --nnodes=1 "
            f"--nproc-per-node {args.gpus} --master-addr 127.0.0.1 "
            "bench.py ..."
        )
    n_gpus = world_size if distributed else args.gpus

    t.cuda.set_device(local_rank)
    device = t.device(f"cuda:{local_rank}")
    t.manual_seed(42 + rank)
    # MIOpen kernel selection, A/B'd on hardware: exhaustive find
    # (benchmark=True) spends ~2.5 s/candidate on naive-conv trials
    # during warmup but lands 1.2 ms igemm kernels for conv2/conv3;
    # immediate-mode FAST find starts instantly but picks 30x-slower
    # kernels (292.9 vs 10.7 ms/step). Exhaustive is the only sane
    # default; warmup absorbs the find phase before the timed window.
    t.backends.cudnn.benchmark = (
        os.environ.get("MACHIN_MIOPEN_BENCHMARK", "1") == "1"
    )

    pipeline = None
    if not args.learner_only:
        n_actors = args.actors
        if n_actors <= 0:
            cpus = os.cpu_count() or 16
            n_actors = max(2, min(8, cpus // (2 * world_size)))
        pipeline = RingPipeline(
            device, args.unroll, args.env_batch, args.actions,
            actors=n_actors, slots=args.slots,
            pool_segments=args.pool_segments,
        )

    # measured: graph replay is NET-NEGATIVE here (3.70 vs 3.86 M
    # samples/s learner-only) — the step is GPU-busy-bound, not
    # launch-gap-bound — so capture stays opt-in
    use_graph = args.graph and not distributed and not args.no_graph
    bench = ImpalaLearnerBench(
        device=device,
        unroll=args.unroll,
        env_batch=args.env_batch,
        action_num=args.actions,
        distributed=distributed,
        capturable=use_graph,
        fused_stem=not args.no_fused_stem,
        bucket_mb=args.bucket_mb,
        ddp_reduction=args.ddp_reduction,
        pipeline=pipeline,
    )

    if pipeline is not None:
        pipeline.start()
    if use_graph:
        try:
            bench.capture_graph()
            print("hipGraph capture: ON", file=sys.stderr)
        except Exception as e:  # noqa: BLE001 - fall back to eager
            print(f"hipGraph capture failed ({e}); eager mode",
                  file=sys.stderr)
            bench._graph = None
    for _ in range(args.warmup):
        bench.step()
    if distributed:
        dist.barrier()
    t.cuda.synchronize()
    if pipeline is not None:
        pipeline.reset_env_steps()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        bench.step()
    t.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    env_steps = pipeline.env_steps() if pipeline is not None else 0
    if distributed:
        et = t.tensor([elapsed], device=device)
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
        es = t.tensor([float(env_steps)], device=device)
        dist.all_reduce(es, op=dist.ReduceOp.SUM)
        dist.barrier()
        elapsed = float(et.item())
        env_steps = int(es.item())
    if pipeline is not None:
        pipeline.stop()

    samples_per_step = args.unroll * args.env_batch
    value = samples_per_step * args.steps * n_gpus / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "impala_learner_samples_per_sec",
                    "value": value,
                    "unit": "samples/s",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": ms_per_step,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16",
                    "data": "synthetic",
                    "env_steps_per_sec": (
                        env_steps / elapsed if pipeline is not None
                        else None
                    ),
                    "pipeline": (
                        "learner_only" if pipeline is None
                        else "actors+ring+hbm_pool"
                    ),
                    "config": {
                        "model": "impala_nature_cnn_atari",
                        "global_batch": samples_per_step * n_gpus,
                        "seq_len": args.unroll,
                        "parallelism": f"dp{n_gpus}",
                    },
                }
            )
        )
    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
