"""Distributed world tests: every collective + rpc primitive at world
size 3 over real gloo + the TCP control plane (reference analog:
test/parallel/distributed/test_world.py)."""
import pytest
import torch as t

from util_run_multi import run_multi


class TestCollectiveGroup:
    def test_all_reduce(self):
        def fn(rank, world):
            group = world.create_collective_group([0, 1, 2])
            x = t.full((4,), float(rank + 1))
            group.all_reduce(x)
            group.destroy()
            return x.tolist()

        results = run_multi(fn)
        for r in results:
            assert r == [6.0] * 4

    def test_broadcast_and_reduce(self):
        def fn(rank, world):
            group = world.create_collective_group([0, 1, 2])
            x = t.full((2,), float(rank))
            group.broadcast(x, src=2)
            bval = x[0].item()
            y = t.full((2,), 1.0)
            group.reduce(y, dst=0)
            group.barrier()
            group.destroy()
            return bval, y[0].item()

        results = run_multi(fn)
        assert [r[0] for r in results] == [2.0, 2.0, 2.0]
        assert results[0][1] == 3.0

    def test_gather_scatter_allgather(self):
        def fn(rank, world):
            group = world.create_collective_group([0, 1, 2])
            x = t.full((2,), float(rank))
            out = [t.zeros(2) for _ in range(3)]
            group.all_gather(out, x)
            ag = [o[0].item() for o in out]
            # scatter
            y = t.zeros(2)
            if rank == 0:
                group.scatter(
                    y, [t.full((2,), float(10 + i)) for i in range(3)], src=0
                )
            else:
                group.scatter(y, None, src=0)
            group.destroy()
            return ag, y[0].item()

        results = run_multi(fn)
        for i, (ag, sc) in enumerate(results):
            assert ag == [0.0, 1.0, 2.0]
            assert sc == 10.0 + i

    def test_send_recv(self):
        def fn(rank, world):
            group = world.create_collective_group([0, 1, 2])
            if rank == 0:
                x = t.full((3,), 42.0)
                group.send(x, dst=1)
                out = 0.0
            elif rank == 1:
                x = t.zeros(3)
                group.recv(x, src=0)
                out = x[0].item()
            else:
                out = -1.0
            group.barrier()
            group.destroy()
            return out

        results = run_multi(fn)
        assert results[1] == 42.0


class TestRpcGroup:
    def test_rpc_sync_async_remote(self):
        def fn(rank, world):
            group = world.create_rpc_group("g", ["0", "1", "2"])
            group.barrier()
            out = {}
            if rank == 0:
                out["sync"] = group.rpc_sync(
                    "1", lambda a, b: a + b, args=(1, 2)
                )
                out["async"] = group.rpc_async(
                    "2", lambda: "hello"
                ).wait()
                out["remote"] = group.remote(
                    "1", lambda x: x * 3, args=(7,)
                ).to_here()
            group.barrier()
            return out

        results = run_multi(fn)
        assert results[0] == {"sync": 3, "async": "hello", "remote": 21}

    def test_rpc_with_tensors(self):
        def fn(rank, world):
            group = world.create_rpc_group("g", ["0", "1", "2"])
            group.barrier()
            out = None
            if rank == 2:
                out = group.rpc_sync(
                    "0", lambda x: (x * 2).sum().item(), args=(t.ones(5),)
                )
            group.barrier()
            return out

        results = run_multi(fn)
        assert results[2] == 10.0

    def test_pair_get_paired(self):
        def fn(rank, world):
            group = world.create_rpc_group("g", ["0", "1", "2"])
            if rank == 1:
                group.pair("model_version", 7)
            group.barrier()
            val = group.get_paired("model_version").to_here()
            paired = group.is_paired("model_version")
            missing = group.is_paired("nonexistent")
            group.barrier()
            if rank == 1:
                group.unpair("model_version")
            group.barrier()
            gone = group.is_paired("model_version")
            group.barrier()
            return val, paired, missing, gone

        results = run_multi(fn)
        for r in results:
            assert r == (7, True, False, False)

    def test_pair_duplicate_raises(self):
        def fn(rank, world):
            group = world.create_rpc_group("g", ["0", "1", "2"])
            group.barrier()
            err = False
            if rank == 0:
                group.pair("k", 1)
                try:
                    group.pair("k", 2)
                except RuntimeError:
                    err = True
            group.barrier()
            return err

        results = run_multi(fn)
        assert results[0] is True

    def test_services(self):
        def fn(rank, world):
            group = world.create_rpc_group("g", ["0", "1", "2"])
            if rank == 0:
                state = {"count": 0}

                def counter(delta):
                    state["count"] += delta
                    return state["count"]

                group.register("counter", counter)
            group.barrier()
            r1 = group.registered_sync("counter", args=(1,))
            group.barrier()
            r2 = group.registered_async("counter", args=(10,)).wait()
            group.barrier()
            r3 = group.registered_remote("counter", args=(0,)).to_here()
            group.barrier()
            assert group.is_registered("counter")
            return r3

        results = run_multi(fn)
        # all three ranks added 1 then 10: final = 33 for everyone
        assert all(r == 33 for r in results)

    def test_barrier_sequencing(self):
        def fn(rank, world):
            import time

            group = world.create_rpc_group("g", ["0", "1", "2"])
            time.sleep(rank * 0.1)
            t0 = time.monotonic()
            group.barrier()
            waited = time.monotonic() - t0
            group.barrier()
            return waited

        results = run_multi(fn)
        # rank 0 should have waited ~0.2s for rank 2
        assert results[0] > 0.1

    def test_non_member_rejected(self):
        def fn(rank, world):
            group = world.create_rpc_group("sub", ["0", "1"])
            err = False
            if rank == 0:
                try:
                    group.rpc_sync("2", lambda: 1)
                except RuntimeError:
                    err = True
            return err

        results = run_multi(fn)
        assert results[0] is True


class TestWorldBasics:
    def test_maps_and_names(self):
        def fn(rank, world):
            return (
                world.get_members(),
                world.name,
                world.rank_name_map[rank],
            )

        results = run_multi(fn, names=["alpha", "beta", "gamma"])
        assert results[0][0] == ["alpha", "beta", "gamma"]
        assert results[1][1] == "beta"


class TestPeerLiveness:
    def test_check_peers(self):
        def fn(rank, world):
            status = world.check_peers()
            return all(status.values()) and len(status) == 3

        assert all(run_multi(fn))


class TestAsyncExecutorBurst:
    def test_many_concurrent_async_calls(self):
        """100 registered_async calls in flight complete on the shared
        executor (round-1 weak #9: thread-per-call churned at APEX
        fan-out rates)."""
        def fn(rank, world):
            group = world.create_rpc_group("burst", ["0", "1", "2"])
            group.register(f"echo_{rank}", lambda x: x * 2)
            group.barrier()
            futures = [
                group.registered_async(
                    f"echo_{(rank + 1) % 3}", args=(i,)
                )
                for i in range(100)
            ]
            results = [f.wait() for f in futures]
            assert results == [i * 2 for i in range(100)]
            group.barrier()
            return True

        assert all(run_multi(fn))
