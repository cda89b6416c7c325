"""Parallel environment execution: in-process and subprocess pools.

Parity target: reference ``machin/env/wrappers/openai_gym.py``:
``ParallelWrapperDummy`` (:24 — sequential for-loop over envs in the
caller) and ``ParallelWrapperSubProc`` (:176-231 — one process per
env, dill-serialized creators, command pipes + shared result queue).
Works with any gym-API environment, including the built-in
machin_amd.env.envs classic-control envs (no gym dependency).
"""
from typing import Any, Callable, List

from ...parallel.pickle import dumps, loads
from ...parallel.process import Process
from .base import ParallelWrapperBase


class ParallelWrapperDummy(ParallelWrapperBase):
    """Run N environments sequentially in the calling process."""

    def __init__(self, env_creators: List[Callable]):
        super().__init__()
        self._envs = [c(i) for i, c in enumerate(env_creators)]
        self._terminal = [False] * len(self._envs)

    def reset(self, idx=None) -> List[Any]:
        idx = self._resolve(idx)
        out = []
        for i in idx:
            self._terminal[i] = False
            out.append(self._envs[i].reset())
        return out

    def step(self, action: List[Any], idx=None):
        idx = self._resolve(idx)
        if len(action) != len(idx):
            raise ValueError("One action per selected environment required.")
        obs, rew, done, info = [], [], [], []
        for a, i in zip(action, idx):
            if self._terminal[i]:
                raise RuntimeError(
                    f"Environment {i} is terminal; reset it first."
                )
            o, r, d, inf = self._envs[i].step(a)
            self._terminal[i] = d
            obs.append(o)
            rew.append(r)
            done.append(d)
            info.append(inf)
        return obs, rew, done, info

    def seed(self, seed=None) -> List[int]:
        if seed is None or isinstance(seed, int):
            seed = [
                (seed or 0) + i for i in range(len(self._envs))
            ]
        for e, s in zip(self._envs, seed):
            e.seed(s)
        return list(seed)

    def render(self, idx=None, *args, **kwargs):
        idx = self._resolve(idx)
        return [self._envs[i].render(*args, **kwargs) for i in idx]

    def close(self):
        for e in self._envs:
            e.close()

    def active(self) -> List[int]:
        return [i for i, d in enumerate(self._terminal) if not d]

    def size(self) -> int:
        return len(self._envs)

    @property
    def action_space(self):
        return self._envs[0].action_space

    @property
    def observation_space(self):
        return self._envs[0].observation_space

    def _resolve(self, idx):
        if idx is None:
            return list(range(len(self._envs)))
        if isinstance(idx, int):
            return [idx]
        return list(idx)


def _subproc_worker(env_creator_bytes, index, conn):
    env = loads(env_creator_bytes)(index)
    try:
        while True:
            cmd, payload = conn.recv()
            if cmd == "reset":
                conn.send(("ok", env.reset()))
            elif cmd == "step":
                conn.send(("ok", env.step(payload)))
            elif cmd == "seed":
                conn.send(("ok", env.seed(payload)))
            elif cmd == "render":
                conn.send(("ok", env.render(**payload)))
            elif cmd == "close":
                env.close()
                conn.send(("ok", None))
                break
    except (EOFError, KeyboardInterrupt):
        pass


class ParallelWrapperSubProc(ParallelWrapperBase):
    """One OS process per environment; commands go over per-env duplex
    pipes (simulation stays on the host CPU farm; MI355X learners
    consume the batched results)."""

    def __init__(self, env_creators: List[Callable]):
        super().__init__()
        import multiprocessing as mp

        ctx = mp.get_context("spawn")
        self._conns = []
        self._procs = []
        for i, creator in enumerate(env_creators):
            parent, child = ctx.Pipe()
            p = Process(
                target=_subproc_worker,
                args=(dumps(creator, recurse=True), i, child),
                ctx=ctx,
                daemon=True,
            )
            p.start()
            self._conns.append(parent)
            self._procs.append(p)
        self._terminal = [False] * len(env_creators)
        # probe one env for spaces
        self._conns[0].send(("seed", 0))
        self._conns[0].recv()
        self._action_space = None
        self._observation_space = None

    def _call(self, idx: List[int], cmd: str, payloads: List[Any]):
        for i, payload in zip(idx, payloads):
            self._conns[i].send((cmd, payload))
        out = []
        for i in idx:
            self._watch(i)
            status, value = self._conns[i].recv()
            out.append(value)
        return out

    def _watch(self, i):
        if not self._procs[i].is_alive():
            self._procs[i].watch()
            raise RuntimeError(f"Environment process {i} died.")

    def reset(self, idx=None):
        idx = self._resolve(idx)
        for i in idx:
            self._terminal[i] = False
        return self._call(idx, "reset", [None] * len(idx))

    def step(self, action, idx=None):
        idx = self._resolve(idx)
        if len(action) != len(idx):
            raise ValueError("One action per selected environment required.")
        for i in idx:
            if self._terminal[i]:
                raise RuntimeError(
                    f"Environment {i} is terminal; reset it first."
                )
        results = self._call(idx, "step", list(action))
        obs, rew, done, info = [], [], [], []
        for i, (o, r, d, inf) in zip(idx, results):
            self._terminal[i] = d
            obs.append(o)
            rew.append(r)
            done.append(d)
            info.append(inf)
        return obs, rew, done, info

    def seed(self, seed=None):
        if seed is None or isinstance(seed, int):
            seed = [(seed or 0) + i for i in range(self.size())]
        self._call(list(range(self.size())), "seed", list(seed))
        return list(seed)

    def render(self, idx=None, *args, **kwargs):
        idx = self._resolve(idx)
        return self._call(idx, "render", [kwargs] * len(idx))

    def close(self):
        try:
            self._call(
                list(range(self.size())), "close", [None] * self.size()
            )
        except (RuntimeError, EOFError, OSError):
            pass
        for p in self._procs:
            p.join(timeout=5)
            if p.is_alive():
                p.terminate()

    def active(self):
        return [i for i, d in enumerate(self._terminal) if not d]

    def size(self):
        return len(self._procs)

    @property
    def action_space(self):
        raise NotImplementedError(
            "Query spaces on a local env instance; subprocess envs do "
            "not proxy space objects."
        )

    @property
    def observation_space(self):
        raise NotImplementedError(
            "Query spaces on a local env instance; subprocess envs do "
            "not proxy space objects."
        )

    def _resolve(self, idx):
        if idx is None:
            return list(range(self.size()))
        if isinstance(idx, int):
            return [idx]
        return list(idx)
