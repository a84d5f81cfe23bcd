from .state import DistributedState, aggregate_metrics_across_ranks, barrier
from .ddp import GradBucketer, wrap_ddp, allreduce_gradients, broadcast_module
from .population_runtime import DistributedPopulation

__all__ = [
    "aggregate_metrics_across_ranks",
    "DistributedState",
    "barrier",
    "GradBucketer",
    "wrap_ddp",
    "allreduce_gradients",
    "broadcast_module",
    "DistributedPopulation",
]
