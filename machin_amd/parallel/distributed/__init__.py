from .world import (
    CollectiveGroup,
    RpcGroup,
    World,
    get_cur_name,
    get_cur_rank,
    get_world,
    is_world_initialized,
)

__all__ = [
    "World",
    "CollectiveGroup",
    "RpcGroup",
    "get_world",
    "get_cur_rank",
    "get_cur_name",
    "is_world_initialized",
]
