from .context import (  # noqa: F401
    DistContext,
    get_context,
    shard_slices,
    spawn_ranks,
)

__all__ = ["DistContext", "get_context", "shard_slices", "spawn_ranks"]
