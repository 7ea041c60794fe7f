from .comm import (  # noqa: F401
    get_tp_group,
    get_tp_rank,
    get_tp_world_size,
    init_tp,
    destroy_tp,
    tp_all_reduce,
    tp_all_gather,
    tp_broadcast_object,
)
