from .dist import (
    DistInfo,
    GradReducer,
    all_reduce_mean_,
    all_reduce_sum_,
    broadcast_module,
    cleanup,
    init_distributed,
)
