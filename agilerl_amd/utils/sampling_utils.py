"""Logits processing for sampling (top-k / top-p / temperature).

Reference parity: ``agilerl/utils/sampling_utils.py`` (used by the legacy
ILQL/BC_LM decode stack; the HF path uses transformers' own processors).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

__all__ = ["process_logits", "top_k_logits", "top_p_logits", "sample_from_logits"]


def top_k_logits(logits: torch.Tensor, k: int) -> torch.Tensor:
    if k <= 0 or k >= logits.shape[-1]:
        return logits
    v, _ = torch.topk(logits, k, dim=-1)
    return logits.masked_fill(logits < v[..., -1:], float("-inf"))


def top_p_logits(logits: torch.Tensor, p: float) -> torch.Tensor:
    if p >= 1.0:
        return logits
    sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
    cum = torch.softmax(sorted_logits, dim=-1).cumsum(dim=-1)
    cut = cum > p
    cut[..., 1:] = cut[..., :-1].clone()
    cut[..., 0] = False
    remove = cut.scatter(-1, sorted_idx, cut)
    return logits.masked_fill(remove, float("-inf"))


def process_logits(
    logits: torch.Tensor,
    temperature: float = 1.0,
    top_k: Optional[int] = None,
    top_p: Optional[float] = None,
) -> torch.Tensor:
    logits = logits / max(temperature, 1e-6)
    if top_k is not None:
        logits = top_k_logits(logits, top_k)
    if top_p is not None:
        logits = top_p_logits(logits, top_p)
    return logits


def sample_from_logits(logits: torch.Tensor, **kwargs) -> torch.Tensor:
    probs = F.softmax(process_logits(logits, **kwargs), dim=-1)
    return torch.multinomial(probs.reshape(-1, probs.shape[-1]), 1).reshape(
        *probs.shape[:-1], 1
    )
