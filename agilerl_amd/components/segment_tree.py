"""Device-resident segment trees for prioritized replay.

Reference parity: ``agilerl/components/segment_tree.py`` (numpy, CPU-only;
batched update :129 and vectorized retrieve :201).  New design: the trees
are flat torch tensors so they live wherever the replay storage lives —
including HBM on MI355X, where batched updates and the sampling descent run
as vectorized torch ops (and dispatch to a HIP descent kernel via
``agilerl_amd.ops`` when built), with **no host round-trips** in the
sample path (SURVEY hard-part #4).
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from ..ops.backend import extension, use_hip

__all__ = ["SumSegmentTree", "MinSegmentTree"]

_OP_SUM, _OP_MIN = 0, 1


class _SegmentTree:
    neutral: float = 0.0
    _hip_op: int = _OP_SUM

    def __init__(self, capacity: int, device: str = "cpu"):
        # round up to a power of two for a perfect binary tree
        self.capacity = 1 << (capacity - 1).bit_length()
        self.device = device
        self.tree = torch.full(
            (2 * self.capacity,), self.neutral, dtype=torch.float32, device=device
        )

    def _combine(self, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:  # pragma: no cover
        raise NotImplementedError

    @torch.no_grad()
    def update(self, idx: torch.Tensor, values: torch.Tensor) -> None:
        """Batched leaf update + path propagation to the root.

        Duplicate indices take the last value (scatter semantics).

        GPU path: one HIP launch (``segtree_update``) — scatter + per-level
        barriered propagation inside a single workgroup (csrc/rl_ops.hip).
        """
        idx = idx.to(self.device).long()
        values = values.to(self.device).float()
        ext = extension()
        if use_hip(self.tree) and ext is not None and hasattr(ext, "segtree_update"):
            ext.segtree_update(
                self.tree, idx.contiguous(), values.contiguous(), self._hip_op
            )
            return
        idx = idx + self.capacity
        self.tree[idx] = values
        # propagate: level by level, recompute affected parents up to the root
        nodes = torch.unique(idx >> 1)
        while True:
            nodes = nodes[nodes >= 1]
            if nodes.numel() == 0:
                break
            self.tree[nodes] = self._combine(self.tree[2 * nodes], self.tree[2 * nodes + 1])
            nodes = torch.unique(nodes >> 1)

    def get(self, idx: torch.Tensor) -> torch.Tensor:
        return self.tree[idx.to(self.device).long() + self.capacity]

    @property
    def root(self) -> torch.Tensor:
        return self.tree[1]


class SumSegmentTree(_SegmentTree):
    neutral = 0.0
    _hip_op = _OP_SUM

    def _combine(self, a, b):
        return a + b

    @torch.no_grad()
    def retrieve(self, prefix: torch.Tensor) -> torch.Tensor:
        """Batched prefix-sum descent: for each p find leaf i with
        cumsum[:i] <= p < cumsum[:i+1].

        GPU path: one lane per prefix with the top tree levels staged in
        LDS (``segtree_retrieve``, csrc/rl_ops.hip); CPU path is a
        vectorized O(B log N) torch descent."""
        prefix = prefix.to(self.device).float()
        ext = extension()
        if use_hip(self.tree) and ext is not None and hasattr(ext, "segtree_retrieve"):
            return ext.segtree_retrieve(self.tree, prefix.contiguous())
        prefix = prefix.clone()
        idx = torch.ones_like(prefix, dtype=torch.long)
        depth = int(math.log2(self.capacity))
        for _ in range(depth):
            left = 2 * idx
            left_val = self.tree[left]
            go_right = prefix >= left_val
            prefix = torch.where(go_right, prefix - left_val, prefix)
            idx = torch.where(go_right, left + 1, left)
        return idx - self.capacity

    def sum(self, start: int = 0, end: Optional[int] = None) -> float:
        if start == 0 and (end is None or end >= self.capacity):
            return float(self.root)
        end = self.capacity if end is None else end
        # rare path — host loop over the range decomposition
        res, lo, hi = 0.0, start + self.capacity, end + self.capacity
        while lo < hi:
            if lo & 1:
                res += float(self.tree[lo])
                lo += 1
            if hi & 1:
                hi -= 1
                res += float(self.tree[hi])
            lo >>= 1
            hi >>= 1
        return res


class MinSegmentTree(_SegmentTree):
    neutral = float("inf")
    _hip_op = _OP_MIN

    def _combine(self, a, b):
        return torch.minimum(a, b)

    def min(self) -> float:
        return float(self.root)
