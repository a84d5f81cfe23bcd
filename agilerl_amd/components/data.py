"""Transition containers.

Reference parity: ``agilerl/components/data.py`` (Transition tensorclass
:81, MultiAgentTransition :126).  tensordict is not a dependency here —
transitions are plain dicts of torch tensors (nested for Dict obs spaces),
which keeps the storage layout explicit for pinned-host staging.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

import numpy as np
import torch

__all__ = ["Transition", "to_tensor", "tree_map", "tree_index", "tree_stack"]


def to_tensor(x, dtype: Optional[torch.dtype] = None) -> Any:
    """Array-like / nested dict -> torch tensor(s) (shared memory where possible)."""
    if isinstance(x, dict):
        return {k: to_tensor(v, dtype) for k, v in x.items()}
    if isinstance(x, (tuple, list)) and x and isinstance(x[0], (dict, np.ndarray, torch.Tensor)):
        return type(x)(to_tensor(v, dtype) for v in x)
    if isinstance(x, torch.Tensor):
        return x.to(dtype) if dtype is not None else x
    t = torch.as_tensor(np.asarray(x))
    return t.to(dtype) if dtype is not None else t


def tree_map(fn, tree):
    if isinstance(tree, dict):
        return {k: tree_map(fn, v) for k, v in tree.items()}
    if isinstance(tree, (tuple, list)):
        return type(tree)(tree_map(fn, v) for v in tree)
    return fn(tree)


def tree_index(tree, idx):
    return tree_map(lambda t: t[idx], tree)


def tree_stack(trees, dim=0):
    first = trees[0]
    if isinstance(first, dict):
        return {k: tree_stack([t[k] for t in trees], dim) for k in first}
    if isinstance(first, (tuple, list)):
        return type(first)(tree_stack([t[i] for t in trees], dim) for i in range(len(first)))
    return torch.stack(list(trees), dim=dim)


@dataclass
class Transition:
    """One (possibly env-batched) transition."""

    obs: Any
    action: Any
    reward: Any
    next_obs: Any
    done: Any
    extras: Dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        d = {
            "obs": self.obs,
            "action": self.action,
            "reward": self.reward,
            "next_obs": self.next_obs,
            "done": self.done,
        }
        d.update(self.extras)
        return d
