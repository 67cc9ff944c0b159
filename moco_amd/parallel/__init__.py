from .comm import (
    all_to_all_rows,
    begin_shuffle,
    concat_all_gather,
    get_rank,
    get_world_size,
    is_distributed,
    shuffle_exchange,
    step_permutation,
    unshuffle_exchange,
)

__all__ = [
    "is_distributed",
    "get_rank",
    "get_world_size",
    "concat_all_gather",
    "step_permutation",
    "shuffle_exchange",
    "unshuffle_exchange",
    "all_to_all_rows",
    "begin_shuffle",
]
