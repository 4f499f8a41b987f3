from opsagent_amd.parallel.state import (
    get_tp_group,
    get_tp_rank,
    get_tp_size,
    init_distributed,
    tp_all_gather,
    tp_all_reduce,
)

__all__ = [
    "init_distributed",
    "get_tp_rank",
    "get_tp_size",
    "get_tp_group",
    "tp_all_reduce",
    "tp_all_gather",
]
