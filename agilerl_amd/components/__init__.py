from .data import Transition, to_tensor, tree_map, tree_index, tree_stack
from .segment_tree import SumSegmentTree, MinSegmentTree
from .replay_buffer import ReplayBuffer, MultiStepReplayBuffer, PrioritizedReplayBuffer
from .rollout_buffer import RolloutBuffer
from .sampler import Sampler

__all__ = [
    "Transition",
    "to_tensor",
    "tree_map",
    "tree_index",
    "tree_stack",
    "SumSegmentTree",
    "MinSegmentTree",
    "ReplayBuffer",
    "MultiStepReplayBuffer",
    "PrioritizedReplayBuffer",
    "RolloutBuffer",
    "Sampler",
]
