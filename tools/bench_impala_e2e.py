"""End-to-end IMPALA pipeline benchmark (BASELINE config #5 shape,
one GPU): a host actor farm writes Atari-shaped rollout segments into
the shared-memory RolloutRing; the learner drains them through pinned
staging onto a side stream into an HBM-resident segment pool, and
trains from the freshest data (actor-lag replay, standard IMPALA
practice when the learner outruns the actors).

Reports BOTH halves of the BASELINE metric:
  * env_steps_per_sec  — fresh environment steps ingested,
  * learner_samples_per_sec — samples consumed by gradient updates.

Run on a GPU box:
  python tools/bench_impala_e2e.py [--actors 12] [--seconds 20]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import torch as t
import torch.multiprocessing as mp

T_UNROLL = 20
SPEC = None  # built in main (needs torch types)


def actor_loop(ring, actor_id: int, stop_flag, steps_counter):
    """Synthetic Atari actor: fills rollout segments in shared memory.
    A real actor would run ALE + the policy net (no ROMs without
    network access); here a pre-generated segment library is cycled
    with cheap mutation, so the measured cost is the TRANSPORT
    pipeline (shared-memory writes, slot bookkeeping), not RNG."""
    import os

    t.set_num_threads(1)
    try:
        os.nice(5)  # the learner's host thread has priority
    except OSError:
        pass
    t.manual_seed(actor_id)
    gen = t.Generator().manual_seed(actor_id)
    library = [
        {
            "frames": t.randint(0, 256, (T_UNROLL, 4, 84, 84),
                                dtype=t.uint8, generator=gen),
            "actions": t.randint(0, 6, (T_UNROLL,), generator=gen),
            "behavior_logp": -t.rand(T_UNROLL, generator=gen) * 2.0,
            "rewards": t.rand(T_UNROLL, generator=gen),
            "terminals": t.zeros(T_UNROLL),
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
        slot["frames"][0, 0, 0, 0] = i % 256  # cheap per-segment variation
        slot["actions"].copy_(src["actions"])
        slot["behavior_logp"].copy_(src["behavior_logp"])
        slot["rewards"].copy_(src["rewards"])
        slot["terminals"].copy_(src["terminals"])
        ring.commit(slot_id)
        with steps_counter.get_lock():
            steps_counter.value += T_UNROLL


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--actors", type=int, default=8)
    parser.add_argument("--seconds", type=float, default=20.0)
    parser.add_argument("--env-batch", type=int, default=1024)
    parser.add_argument("--pool-segments", type=int, default=4096)
    parser.add_argument("--slots", type=int, default=256)
    args = parser.parse_args()

    from bench import ImpalaLearnerBench
    from machin_amd.parallel.rollout_ring import RolloutRing

    ctx = mp.get_context("spawn")
    spec = {
        "frames": ((T_UNROLL, 4, 84, 84), t.uint8),
        "actions": ((T_UNROLL,), t.long),
        "behavior_logp": ((T_UNROLL,), t.float32),
        "rewards": ((T_UNROLL,), t.float32),
        "terminals": ((T_UNROLL,), t.float32),
    }
    ring = RolloutRing(slots=args.slots, spec=spec, ctx=ctx)
    stop_flag = t.zeros(1, dtype=t.uint8).share_memory_()
    steps_counter = ctx.Value("q", 0)
    actors = [
        ctx.Process(
            target=actor_loop, args=(ring, a, stop_flag, steps_counter),
            daemon=True,
        )
        for a in range(args.actors)
    ]
    for p in actors:
        p.start()

    dev = t.device("cuda:0")
    t.backends.cudnn.benchmark = True
    bench = ImpalaLearnerBench(
        device=dev, unroll=T_UNROLL, env_batch=args.env_batch,
        pool_size=1,
    )
    # HBM-resident segment pool, filled from the ring
    M = args.pool_segments
    pool = {
        k: t.zeros((M, *shape), dtype=dtype, device=dev)
        for k, (shape, dtype) in spec.items()
    }
    pinned = ring.make_pinned_staging(64)
    registered = ring.host_register()
    print("ring host_register (direct DMA):", registered, file=sys.stderr)
    side = t.cuda.Stream()
    state = {"filled": 0, "write_pos": 0, "stop": False}

    def ingest_once(timeout=0.05):
        idx = ring.drain(max_slots=64, timeout=timeout)
        if not idx:
            return 0
        wp = state["write_pos"]
        n = len(idx)
        with t.cuda.stream(side):
            if registered:
                positions = [(wp + i) % M for i in range(n)]
                ring.upload_slots(idx, pool, positions)
            else:
                batch = ring.gather(idx, dev, pinned=pinned)
                for k, v in batch.items():
                    end = wp + n
                    if end <= M:
                        pool[k][wp:end] = v
                    else:
                        split = M - wp
                        pool[k][wp:] = v[:split]
                        pool[k][: end % M] = v[split:]
        side.synchronize()
        ring.release(idx)
        state["write_pos"] = (wp + n) % M
        state["filled"] = min(state["filled"] + n, M)
        return n

    def ingest_loop():
        # background staging: drains the ring and uploads to HBM on a
        # side stream, fully overlapped with the training thread
        # (host memcpys release the GIL)
        while not state["stop"]:
            ingest_once()

    import threading

    def train_step():
        B = args.env_batch
        seg = t.randint(0, max(state["filled"], 1), (B,), device=dev)
        data = {
            # t-major flattening: the learner views logits as [T, B]
            "frames": pool["frames"].index_select(0, seg)
            .permute(1, 0, 2, 3, 4).reshape(T_UNROLL * B, 4, 84, 84)
            .to(memory_format=t.channels_last),
            "actions": pool["actions"].index_select(0, seg).t()
            .contiguous(),
            "behavior_logp": pool["behavior_logp"].index_select(0, seg)
            .t().contiguous(),
            "rewards": pool["rewards"].index_select(0, seg).t()
            .contiguous(),
            "terminals": pool["terminals"].index_select(0, seg).t()
            .contiguous(),
        }
        return bench._step_body(data)

    # warm up: fill enough segments, run a few steps
    while state["filled"] < args.env_batch:
        ingest_once(timeout=1.0)
    for _ in range(5):
        train_step()
    t.cuda.synchronize()
    ingester = threading.Thread(target=ingest_loop, daemon=True)
    ingester.start()

    with steps_counter.get_lock():
        steps_counter.value = 0
    trained_samples = 0
    steps = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < args.seconds:
        train_step()
        steps += 1
        trained_samples += args.env_batch * T_UNROLL
    t.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    state["stop"] = True
    ingester.join(timeout=5)
    with steps_counter.get_lock():
        env_steps = steps_counter.value
    stop_flag[0] = 1
    for p in actors:
        p.join(timeout=5)
        if p.is_alive():
            p.terminate()

    out = {
        "metric": "impala_e2e",
        "learner_samples_per_sec": trained_samples / elapsed,
        "env_steps_per_sec": env_steps / elapsed,
        "reuse_factor": trained_samples / max(env_steps, 1),
        "ms_per_learner_step": elapsed / steps * 1000,
        "actors": args.actors,
        "env_batch": args.env_batch,
        "unroll": T_UNROLL,
        "dtype": "bf16",
        "data": "synthetic",
    }
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/impala_e2e.json", "w") as f:
        json.dump(out, f, indent=2)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
