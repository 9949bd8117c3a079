from .fanout import (
    FanoutContext,
    all_gather_rows,
    broadcast_tensors,
    init_fanout,
    shard_rows,
)

__all__ = [
    "FanoutContext",
    "init_fanout",
    "broadcast_tensors",
    "shard_rows",
    "all_gather_rows",
]
