"""Server construction helpers.

Parity target: reference ``machin/frame/helpers/servers.py`` —
``model_server_helper`` (:112) and ``grad_server_helper`` (:7):
convenience constructors every process calls collectively; the
designated server member(s) build the Impl objects, everyone else
receives accessors through value pairing.
"""
from typing import Callable, List

import torch as t

from ...parallel.distributed.world import get_world
from ...parallel.server.param_server import (
    PushPullGradServerImpl,
    PushPullModelServerImpl,
)


def model_server_helper(model_num: int,
                        group_name: str = "model_server_group",
                        members: List[str] = None):
    """Create ``model_num`` push/pull model servers hosted on the first
    member. Returns a tuple of accessors (one per model)."""
    world = get_world()
    if world is None:
        raise RuntimeError("World must be initialized first.")
    members = members or world.get_members()
    group = world.create_rpc_group(group_name, world.get_members())
    if world.name == members[0]:
        for i in range(model_num):
            PushPullModelServerImpl(f"model_server_{i}", group)
    group.barrier()
    servers = tuple(
        group.get_paired(f"model_server_{i}").to_here()
        for i in range(model_num)
    )
    group.barrier()
    return servers


def grad_server_helper(
    model_creators: List[Callable],
    group_name: str = "grad_server_group",
    members: List[str] = None,
    optimizer=t.optim.Adam,
    learning_rate=1e-3,
    optimizer_kwargs: List[dict] = None,
    reduce_method: str = "sum",
    reduce_device="cpu",
    reduce_batch_size: int = 4,
    max_queue_size: int = 64,
):
    """Create one gradient-reduction server per model creator. The
    first member manages models/optimizers; every member hosts a
    secondary reducer. Returns a tuple of accessors."""
    world = get_world()
    if world is None:
        raise RuntimeError("World must be initialized first.")
    members = members or world.get_members()
    group = world.create_rpc_group(group_name, world.get_members())
    lr = (
        learning_rate
        if isinstance(learning_rate, (list, tuple))
        else [learning_rate] * len(model_creators)
    )
    optimizer_kwargs = optimizer_kwargs or [{}] * len(model_creators)
    if world.name in members:
        for i, creator in enumerate(model_creators):
            impl = PushPullGradServerImpl(
                f"grad_server_{i}",
                group,
                reduce_method=reduce_method,
                reduce_device=reduce_device,
                reduce_batch_size=reduce_batch_size,
                max_queue_size=max_queue_size,
                reducer_members=members,
            )
            if world.name == members[0]:
                model = creator()
                impl.manage_model(
                    model,
                    optimizer(
                        model.parameters(), lr=lr[i], **optimizer_kwargs[i]
                    ),
                )
    group.barrier()
    servers = tuple(
        group.get_paired(f"grad_server_{i}").to_here()
        for i in range(len(model_creators))
    )
    group.barrier()
    return servers
