"""RolloutRing: actor processes write shared-memory slots; the
consumer drains, gathers and recycles."""
import torch as t
import torch.multiprocessing as mp

from machin_amd.parallel.rollout_ring import RolloutRing

SPEC = {
    "frames": ((5, 8), t.uint8),
    "rewards": ((5,), t.float32),
}


def _actor(ring: RolloutRing, actor_id: int, n_segments: int):
    t.manual_seed(actor_id)
    for _ in range(n_segments):
        slot_id = ring.acquire(timeout=30)
        slot = ring.slot(slot_id)
        slot["frames"].fill_(actor_id + 1)
        slot["rewards"].fill_(float(actor_id + 1))
        ring.commit(slot_id)


class TestRolloutRing:
    def test_end_to_end(self):
        ctx = mp.get_context("spawn")
        ring = RolloutRing(slots=8, spec=SPEC, ctx=ctx)
        n_actors, per_actor = 3, 4
        procs = [
            ctx.Process(target=_actor, args=(ring, a, per_actor))
            for a in range(n_actors)
        ]
        for p in procs:
            p.start()
        staging = ring.make_pinned_staging(8) if t.cuda.is_available() \
            else None
        consumed = 0
        seen_values = set()
        while consumed < n_actors * per_actor:
            idx = ring.drain(max_slots=4, timeout=10)
            assert idx, "ring drain timed out"
            batch = ring.gather(idx, t.device("cpu"), pinned=staging)
            assert batch["frames"].shape == (len(idx), 5, 8)
            for row in batch["frames"]:
                v = int(row[0, 0])
                assert v in (1, 2, 3)
                seen_values.add(v)
            # reward and frames of the same slot agree
            assert t.allclose(
                batch["rewards"][:, 0],
                batch["frames"][:, 0, 0].float(),
            )
            ring.release(idx)
            consumed += len(idx)
        for p in procs:
            p.join(timeout=10)
        assert seen_values == {1, 2, 3}

    def test_backpressure(self):
        """Actors block on acquire when the consumer lags; no data is
        overwritten while held."""
        ctx = mp.get_context("spawn")
        ring = RolloutRing(slots=2, spec=SPEC, ctx=ctx)
        p = ctx.Process(target=_actor, args=(ring, 0, 4))
        p.start()
        got = []
        while len(got) < 4:
            idx = ring.drain(max_slots=1, timeout=15)
            assert idx
            batch = ring.gather(idx, t.device("cpu"))
            assert int(batch["frames"][0, 0, 0]) == 1
            ring.release(idx)
            got.extend(idx)
        p.join(timeout=10)
