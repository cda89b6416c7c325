"""Tests for machin_amd.parallel: processes, threads, queues, pickle,
events, pools."""
import multiprocessing as mp
import time

import pytest
import torch as t

from machin_amd.parallel.event import AndEvent, Event, OrEvent
from machin_amd.parallel.pickle import dumps, loads
from machin_amd.parallel.pool import (
    CtxPool,
    CtxThreadPool,
    P2PPool,
    Pool,
    ThreadPool,
)
from machin_amd.parallel.process import Process, ProcessException
from machin_amd.parallel.queue import SimpleP2PQueue, SimpleQueue
from machin_amd.parallel.thread import Thread, ThreadException


def _fail():
    raise RuntimeError("boom")


class TestProcess:
    def test_ok(self):
        p = Process(target=time.sleep, args=(0.01,))
        p.start()
        p.join()
        p.watch()

    def test_exception_piped(self):
        p = Process(target=_fail)
        p.start()
        p.join()
        with pytest.raises(ProcessException, match="boom"):
            p.watch()


class TestThread:
    def test_exception(self):
        th = Thread(target=_fail)
        th.start()
        th.join()
        with pytest.raises(ThreadException, match="boom"):
            th.watch()


class TestPickle:
    def test_lambda(self):
        f = loads(dumps(lambda x: x * 2))
        assert f(21) == 42

    def test_closure(self):
        y = 10
        f = loads(dumps(lambda x: x + y, recurse=True))
        assert f(1) == 11

    def test_tensor_copy(self):
        x = t.rand(3, 3)
        y = loads(dumps(x))
        assert t.allclose(x, y)
        y += 1
        assert not t.allclose(x, y)

    def test_tensor_shared(self):
        x = t.rand(3, 3)
        x.share_memory_()
        y = loads(dumps(x, copy_tensor=False))
        y += 1
        assert t.allclose(x, y)  # same storage


class TestQueues:
    def test_simple_queue(self):
        q = SimpleQueue()
        q.put({"a": t.ones(2)})
        out = q.get(timeout=1)
        assert out["a"].sum() == 2
        with pytest.raises(Exception):
            q.get(timeout=0.05)

    def test_p2p_queue(self):
        q = SimpleP2PQueue()
        q.put([1, 2, 3])
        assert q.get(timeout=1) == [1, 2, 3]
        assert q.empty()

    def test_queue_across_process(self):
        q = SimpleQueue(ctx=mp.get_context("fork"))

        def child(qq):
            qq.put("from child")

        p = Process(target=child, args=(q,))
        p.start()
        assert q.get(timeout=5) == "from child"
        p.join()


class TestEvents:
    def test_or(self):
        a, b = Event(), Event()
        oe = OrEvent(a, b)
        assert not oe.is_set()
        a.set()
        assert oe.is_set()
        a.clear()
        assert not oe.is_set()

    def test_and(self):
        a, b = Event(), Event()
        ae = AndEvent(a, b)
        a.set()
        assert not ae.is_set()
        b.set()
        assert ae.is_set()

    def test_nested(self):
        a, b, c = Event(), Event(), Event()
        e = OrEvent(AndEvent(a, b), c)
        c.set()
        assert e.is_set()
        c.clear()
        a.set(); b.set()
        assert e.is_set()

    def test_direct_set_forbidden(self):
        with pytest.raises(RuntimeError):
            OrEvent(Event(), Event()).set()


class TestPools:
    @pytest.mark.parametrize("pool_cls", [Pool, P2PPool])
    def test_map_lambda(self, pool_cls):
        with pool_cls(processes=2) as pool:
            out = pool.map(lambda x: x * 2, [1, 2, 3, 4])
            assert out == [2, 4, 6, 8]

    def test_apply_and_async(self):
        with Pool(processes=2) as pool:
            assert pool.apply(lambda a, b: a + b, (1, 2)) == 3
            r = pool.apply_async(lambda: 7)
            assert r.get(timeout=10) == 7
            assert r.successful()

    def test_starmap(self):
        with Pool(processes=2) as pool:
            assert pool.starmap(lambda a, b: a * b, [(2, 3), (4, 5)]) == [6, 20]

    def test_imap_unordered(self):
        with Pool(processes=2) as pool:
            out = sorted(pool.imap_unordered(lambda x: x + 1, [1, 2, 3]))
            assert out == [2, 3, 4]

    def test_worker_exception(self):
        with Pool(processes=1) as pool:
            with pytest.raises(RuntimeError, match="boom"):
                pool.apply(_fail)

    def test_tensor_passing(self):
        with Pool(processes=2) as pool:
            out = pool.map(lambda x: x.sum().item(), [t.ones(4), t.ones(2)])
            assert out == [4.0, 2.0]

    def test_closure_capture(self):
        bias = 100
        with Pool(processes=2) as pool:
            assert pool.map(lambda x: x + bias, [1, 2]) == [101, 102]

    def test_ctx_pool(self):
        with CtxPool(processes=2, worker_contexts=["a", "b"]) as pool:
            out = set(pool.map(lambda ctx, x: f"{ctx}{x}", [1, 2, 3, 4]))
            assert out <= {"a1", "a2", "a3", "a4", "b1", "b2", "b3", "b4"}

    def test_thread_pool(self):
        with ThreadPool(processes=2) as pool:
            assert pool.map(lambda x: x ** 2, [1, 2, 3]) == [1, 4, 9]

    def test_ctx_thread_pool(self):
        with CtxThreadPool(processes=2) as pool:
            out = pool.map(lambda ctx, x: x + ctx, [10, 10])
            assert all(v in (10, 11) for v in out)


class TestExceptionTransport:
    def test_pickled_traceback(self):
        import pickle

        from machin_amd.parallel.exception import (
            ExceptionWithTraceback,
            RemoteTraceback,
        )

        try:
            raise ValueError("worker boom")
        except ValueError as e:
            data = pickle.dumps(ExceptionWithTraceback(e))
        exc = pickle.loads(data)
        assert isinstance(exc, ValueError)
        assert isinstance(exc.__cause__, RemoteTraceback)
        assert "worker boom" in str(exc.__cause__)


class TestMultiP2PQueue:
    def test_round_robin(self):
        from machin_amd.parallel.queue import MultiP2PQueue

        q = MultiP2PQueue(3)
        for i in range(6):
            q.put(i)
        got = sorted(q.get(timeout=1) for _ in range(6))
        assert got == [0, 1, 2, 3, 4, 5]
        # sub-queue access
        q.put("x")
        assert q.get_sub_queue(0).get(timeout=1) == "x"
        q.close()

    def test_get_timeout(self):
        from machin_amd.parallel.queue import Empty, MultiP2PQueue

        q = MultiP2PQueue(2)
        with pytest.raises(Empty):
            q.get(timeout=0.05)
        q.close()


def _inc(x):
    return x + 1


class TestQueueContention:
    def test_simple_queue_contended_get_put(self):
        """Regression: multiprocessing SemLocks treat negative
        timeouts as TRY-ONCE; SimpleQueue used timeout=-1 for
        "blocking" acquire, so contended put/get raised spurious
        Full/Empty and killed pool workers under load."""
        import threading

        from machin_amd.parallel.queue import SimpleQueue

        q = SimpleQueue()
        n_threads, per_thread = 8, 200
        errors = []

        def producer(base):
            try:
                for i in range(per_thread):
                    q.put(base + i)
            except Exception as e:  # noqa: BLE001
                errors.append(e)

        got = []
        got_lock = threading.Lock()

        def consumer():
            try:
                for _ in range(per_thread * n_threads // 4):
                    v = q.get(timeout=30)
                    with got_lock:
                        got.append(v)
            except Exception as e:  # noqa: BLE001
                errors.append(e)

        producers = [
            threading.Thread(target=producer, args=(k * per_thread,))
            for k in range(n_threads)
        ]
        consumers = [threading.Thread(target=consumer) for _ in range(4)]
        for th in producers + consumers:
            th.start()
        for th in producers + consumers:
            th.join(timeout=60)
        assert not errors, errors
        assert len(got) == n_threads * per_thread
        assert len(set(got)) == len(got)

    def test_pool_survives_many_small_tasks(self):
        """2000 chunk-1 tasks through the process pool: no worker
        churn, all results correct (previously workers died on
        contended queue locks)."""
        from machin_amd.parallel.pool import Pool

        p = Pool(processes=4)
        try:
            out = p.map(_inc, range(500))
            assert out == [x + 1 for x in range(500)]
        finally:
            p.terminate()
