from .event import AndEvent, Event, OrEvent
from .exception import ExceptionWithTraceback, RemoteTraceback
from .pickle import Pickler, dumps, loads, mark_static_module
from .pool import CtxPool, CtxThreadPool, P2PPool, Pool, ThreadPool
from .process import Process, ProcessException
from .queue import MultiP2PQueue, SimpleP2PQueue, SimpleQueue
from .thread import Thread, ThreadException
from .util import Finalize

__all__ = [
    "Process",
    "ProcessException",
    "Thread",
    "ThreadException",
    "Pickler",
    "dumps",
    "loads",
    "mark_static_module",
    "SimpleQueue",
    "SimpleP2PQueue",
    "MultiP2PQueue",
    "Pool",
    "P2PPool",
    "CtxPool",
    "ThreadPool",
    "CtxThreadPool",
    "Event",
    "OrEvent",
    "AndEvent",
    "Finalize",
    "ExceptionWithTraceback",
    "RemoteTraceback",
]


def __getattr__(name):
    # heavy / world-touching members load lazily
    if name in ("World", "CollectiveGroup", "RpcGroup", "get_world",
                "get_cur_rank", "get_cur_name"):
        from .distributed import world as _w

        return getattr(_w, name)
    if name in ("GradReducer", "DistributedDataParallel"):
        from . import ddp as _d

        return getattr(_d, name)
    if name == "ModelAssigner":
        from .assigner import ModelAssigner

        return ModelAssigner
    if name == "RolloutRing":
        from .rollout_ring import RolloutRing

        return RolloutRing
    raise AttributeError(name)
