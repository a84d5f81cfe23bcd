"""Logits processing for sampling (top-k / top-p / temperature).

Reference parity: ``agilerl/utils/sampling_utils.py`` (top-k/top-p logits
processing used by the legacy ILQL/BC_LM decoding stack).
"""

from __future__ import annotations

from typing import Optional

import torch

__all__ = ["process_logits", "sample_from_logits"]


def process_logits(
    logits: torch.Tensor,
    temperature: float = 1.0,
    top_k: Optional[int] = None,
    top_p: Optional[float] = None,
) -> torch.Tensor:
    """Temperature-scale then mask logits outside the top-k / nucleus set.

    logits: (..., V).  Masked entries become -inf so softmax renormalizes
    over the kept set.
    """
    logits = logits / max(float(temperature), 1e-6)
    if top_k is not None and 0 < top_k < logits.shape[-1]:
        kth = torch.topk(logits, top_k, dim=-1).values[..., -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if top_p is not None and 0.0 < top_p < 1.0:
        sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
        probs = torch.softmax(sorted_logits, dim=-1)
        cum = probs.cumsum(dim=-1)
        # keep the smallest prefix with cumulative mass >= top_p (always
        # keep the first token)
        drop_sorted = cum - probs >= top_p
        drop = torch.zeros_like(drop_sorted).scatter(-1, sorted_idx, drop_sorted)
        logits = logits.masked_fill(drop, float("-inf"))
    return logits


def sample_from_logits(
    logits: torch.Tensor,
    temperature: float = 1.0,
    top_k: Optional[int] = None,
    top_p: Optional[float] = None,
    greedy: bool = False,
) -> torch.Tensor:
    """(..., V) -> (...,) sampled token ids."""
    if greedy:
        return logits.argmax(dim=-1)
    proc = process_logits(logits, temperature, top_k, top_p)
    probs = torch.softmax(proc, dim=-1)
    flat = probs.reshape(-1, probs.shape[-1])
    out = torch.multinomial(flat, 1).squeeze(-1)
    return out.reshape(probs.shape[:-1])
