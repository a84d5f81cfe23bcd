"""Evolvable BERT (encoder-only transformer).

Reference parity: ``agilerl/modules/bert.py:16`` (EvolvableBERT).  Same
block structure as :class:`EvolvableGPT` but bidirectional attention and
a pooled [CLS]-style output head option.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from .gpt import EvolvableGPT

__all__ = ["EvolvableBERT"]


class EvolvableBERT(EvolvableGPT):
    CAUSAL = False

    def __init__(
        self,
        vocab_size: int,
        n_layer: int = 4,
        n_head: int = 4,
        n_embd: int = 128,
        max_positions: int = 512,
        dropout: float = 0.0,
        num_outputs: Optional[int] = None,
        device: str = "cpu",
    ):
        self.num_outputs = num_outputs
        super().__init__(
            vocab_size=vocab_size, n_layer=n_layer, n_head=n_head, n_embd=n_embd,
            max_positions=max_positions, dropout=dropout, device=device,
        )
        if num_outputs is not None:
            self.cls_head = nn.Linear(self.n_embd, num_outputs).to(device)

    def forward(self, idx: torch.Tensor, targets: Optional[torch.Tensor] = None):
        hidden = self.transformer_forward(idx)
        if self.num_outputs is not None and targets is None:
            return self.cls_head(hidden[:, 0])  # pooled first-token output
        return super().forward(idx, targets)

    def recreate_network(self) -> None:
        super().recreate_network()
        if self.num_outputs is not None:
            from .base import preserve_parameters

            new_head = nn.Linear(self.n_embd, self.num_outputs).to(self.device)
            preserve_parameters(self.cls_head, new_head)
            self.cls_head = new_head
