"""KV cache for the legacy EvolvableGPT decode stack.

Reference parity: ``agilerl/utils/cache.py:11`` (Cache container used by
ILQL/BC_LM generation).  Per-layer preallocated K/V rings sized to
``max_positions`` so incremental decode appends in place.
"""

from __future__ import annotations

from typing import List, Tuple

import torch

__all__ = ["Cache"]


class Cache:
    def __init__(self):
        self.keys: List[torch.Tensor] = []
        self.values: List[torch.Tensor] = []
        self.length = 0

    def update(self, layer: int, k: torch.Tensor, v: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """Append (B, H, T_new, D) keys/values for ``layer``; returns the
        full cached (B, H, T_total, D) tensors."""
        if layer >= len(self.keys):
            self.keys.append(k)
            self.values.append(v)
        else:
            self.keys[layer] = torch.cat([self.keys[layer], k], dim=2)
            self.values[layer] = torch.cat([self.values[layer], v], dim=2)
        if layer == 0:
            self.length = self.keys[0].shape[2]
        return self.keys[layer], self.values[layer]

    def reset(self) -> None:
        self.keys.clear()
        self.values.clear()
        self.length = 0

    def trim(self, max_len: int) -> None:
        if self.length <= max_len:
            return
        self.keys = [k[:, :, -max_len:] for k in self.keys]
        self.values = [v[:, :, -max_len:] for v in self.values]
        self.length = max_len
