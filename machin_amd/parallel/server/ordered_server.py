"""Version-ordered key-value server.

Parity target: reference ``machin/parallel/server/ordered_server.py``
(:54-142): ``push(key, value, version, prev_version)`` succeeds only
if ``prev_version`` matches the current tail version; ``pull(key,
version=None)`` returns ``(value, version)`` of the requested (or
newest) version. Accessor / implementation split: the Impl registers
services on an RpcGroup, accessors call them from any member.
"""
import threading
from collections import OrderedDict
from typing import Any

from ..distributed.world import RpcGroup


class OrderedServerBase:
    def push(self, key, value, version, prev_version) -> bool:
        raise NotImplementedError

    def pull(self, key, version=None):
        raise NotImplementedError


class OrderedServerSimple(OrderedServerBase):
    """Accessor: calls the Impl's registered services."""

    def __init__(self, server_name: str, group: RpcGroup):
        self.server_name = server_name
        self.group = group

    def push(self, key, value, version, prev_version) -> bool:
        return self.group.registered_sync(
            self.server_name + "/_push_service",
            args=(key, value, version, prev_version),
        )

    def pull(self, key, version=None):
        return self.group.registered_sync(
            self.server_name + "/_pull_service", args=(key, version)
        )


class OrderedServerSimpleImpl:
    """The actual store; construct on exactly one member."""

    def __init__(self, server_name: str, group: RpcGroup,
                 version_depth: int = 1):
        if version_depth <= 0:
            raise ValueError("version_depth must be positive.")
        self.server_name = server_name
        self.group = group
        self.version_depth = version_depth
        self._data = {}
        self._lock = threading.Lock()
        group.register(server_name + "/_push_service", self._push_service)
        group.register(server_name + "/_pull_service", self._pull_service)
        group.pair(server_name, OrderedServerSimple(server_name, group))

    def _push_service(self, key, value, version, prev_version) -> bool:
        with self._lock:
            chain = self._data.get(key)
            if chain is None:
                chain = OrderedDict()
                self._data[key] = chain
            if chain:
                tail = next(reversed(chain))
                if tail != prev_version:
                    return False
            chain[version] = value
            while len(chain) > self.version_depth:
                chain.popitem(last=False)
            return True

    def _pull_service(self, key, version=None):
        with self._lock:
            chain = self._data.get(key)
            if not chain:
                return None
            if version is None:
                version = next(reversed(chain))
            if version not in chain:
                return None
            return (chain[version], version)
