from .engine import (  # noqa: F401
    init_distributed,
    is_distributed,
    shard_bounds,
    broadcast_array,
    allgather_rows,
    explain_sharded,
)
