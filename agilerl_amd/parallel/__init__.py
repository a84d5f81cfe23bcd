from .state import DistributedState, barrier
from .ddp import GradBucketer, wrap_ddp, allreduce_gradients, broadcast_module
from .population_runtime import DistributedPopulation

__all__ = [
    "DistributedState",
    "barrier",
    "GradBucketer",
    "wrap_ddp",
    "allreduce_gradients",
    "broadcast_module",
    "DistributedPopulation",
]
