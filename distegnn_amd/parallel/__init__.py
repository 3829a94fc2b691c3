from .comm import (
    CapturedAllReduce,
    GradBucket,
    all_reduce_sum_differentiable,
    barrier,
    capture_comm_fallback,
    check_model_parameters,
    destroy,
    fused_weighted_average_reduce,
    global_counts,
    init_distributed,
    is_distributed,
    rank,
    world_size,
)

__all__ = [
    "CapturedAllReduce", "GradBucket", "all_reduce_sum_differentiable",
    "barrier", "capture_comm_fallback", "check_model_parameters", "destroy",
    "fused_weighted_average_reduce", "global_counts", "init_distributed",
    "is_distributed", "rank", "world_size",
]
