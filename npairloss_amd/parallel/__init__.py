from .collectives import (
    world_size,
    rank,
    all_gather_rows,
    reduce_scatter_rows,
    all_reduce_sum,
)

__all__ = [
    "world_size",
    "rank",
    "all_gather_rows",
    "reduce_scatter_rows",
    "all_reduce_sum",
]
