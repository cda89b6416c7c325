"""Pipe-based queues with dill serialization and timeouts.

Parity target: reference ``machin/parallel/queue.py`` (:42-283):
``SimpleQueue`` (multi-producer multi-consumer over one pipe pair with
locks, timeout-capable, plus lockless quick_put/quick_get for
single-user ends), ``SimpleP2PQueue`` (1:1, lockless) and
``MultiP2PQueue`` (round-robin over per-worker P2P queues).
"""
import multiprocessing as mp
import queue as q_exc
import time
from multiprocessing import connection
from typing import Any, List

from .pickle import dumps, loads

Empty = q_exc.Empty
Full = q_exc.Full


class SimpleQueue:
    """Multi-producer multi-consumer pipe queue."""

    def __init__(self, ctx=mp, copy_tensor: bool = True):
        self._reader, self._writer = connection.Pipe(duplex=False)
        self._rlock = ctx.Lock()
        self._wlock = ctx.Lock()
        self.copy_tensor = copy_tensor

    def empty(self) -> bool:
        return not self._reader.poll()

    def put(self, obj: Any, timeout: float = None):
        data = dumps(obj, copy_tensor=self.copy_tensor)
        # NOTE: multiprocessing SemLock treats a NEGATIVE timeout as
        # try-once, not infinite (unlike threading.Lock) — passing -1
        # made contended put/get kill pool workers with spurious
        # Full/Empty under load
        acquired = (
            self._wlock.acquire()
            if timeout is None
            else self._wlock.acquire(timeout=timeout)
        )
        if not acquired:
            raise Full("put timed out")
        try:
            self._writer.send_bytes(data)
        finally:
            self._wlock.release()

    def get(self, timeout: float = None) -> Any:
        deadline = None if timeout is None else time.monotonic() + timeout
        acquired = (
            self._rlock.acquire()
            if timeout is None
            else self._rlock.acquire(timeout=timeout)
        )
        if not acquired:
            raise Empty("get timed out")
        try:
            remain = (
                None if deadline is None else max(deadline - time.monotonic(), 0)
            )
            if remain is not None and not self._reader.poll(remain):
                raise Empty("get timed out")
            data = self._reader.recv_bytes()
        finally:
            self._rlock.release()
        return loads(data)

    def quick_put(self, obj: Any):
        """Lockless put: only safe with a single producer."""
        self._writer.send_bytes(dumps(obj, copy_tensor=self.copy_tensor))

    def quick_get(self, timeout: float = None) -> Any:
        """Lockless get: only safe with a single consumer."""
        if timeout is not None and not self._reader.poll(timeout):
            raise Empty("get timed out")
        return loads(self._reader.recv_bytes())

    def close(self):
        self._reader.close()
        self._writer.close()

    def __getstate__(self):
        return (
            self._reader, self._writer, self._rlock, self._wlock,
            self.copy_tensor,
        )

    def __setstate__(self, state):
        (
            self._reader, self._writer, self._rlock, self._wlock,
            self.copy_tensor,
        ) = state


class SimpleP2PQueue:
    """1-producer 1-consumer queue, no locks."""

    def __init__(self, copy_tensor: bool = True):
        self._reader, self._writer = connection.Pipe(duplex=False)
        self.copy_tensor = copy_tensor

    def empty(self) -> bool:
        return not self._reader.poll()

    def put(self, obj: Any, timeout: float = None):
        self._writer.send_bytes(dumps(obj, copy_tensor=self.copy_tensor))

    def get(self, timeout: float = None) -> Any:
        if timeout is not None and not self._reader.poll(timeout):
            raise Empty("get timed out")
        return loads(self._reader.recv_bytes())

    quick_put = put
    quick_get = get

    def close(self):
        self._reader.close()
        self._writer.close()

    def __getstate__(self):
        return (self._reader, self._writer, self.copy_tensor)

    def __setstate__(self, state):
        self._reader, self._writer, self.copy_tensor = state


class MultiP2PQueue:
    """Fan-out over N P2P queues, round-robin put, polling get."""

    def __init__(self, queue_num: int, copy_tensor: bool = True):
        self.queues: List[SimpleP2PQueue] = [
            SimpleP2PQueue(copy_tensor) for _ in range(queue_num)
        ]
        self._put_idx = 0

    def put(self, obj: Any):
        self.queues[self._put_idx].put(obj)
        self._put_idx = (self._put_idx + 1) % len(self.queues)

    def get(self, timeout: float = None) -> Any:
        """Get from any sub-queue — event-driven multiplexing over
        the pipe readers (the previous 100 µs sleep-poll loop capped
        small-result throughput)."""
        from multiprocessing.connection import wait

        deadline = None if timeout is None else time.monotonic() + timeout
        readers = {q._reader: q for q in self.queues}
        while True:
            remain = (
                None if deadline is None
                else max(deadline - time.monotonic(), 0)
            )
            ready = wait(list(readers), timeout=remain)
            if ready:
                return readers[ready[0]].get()
            if deadline is not None and time.monotonic() >= deadline:
                raise Empty("get timed out")

    def get_sub_queue(self, index: int) -> SimpleP2PQueue:
        return self.queues[index]

    def close(self):
        for q in self.queues:
            q.close()
