"""Padding-free sequence packing.

Reference parity: ``agilerl/utils/llm_packing.py`` (``pack_padded_batch``
:59 -> varlen layout + per-sequence position_ids; ``unpack_logprobs``
:181).  Packing removes pad tokens before the gradient forward so the
fused lm_head path processes only real tokens — on MI355X that means the
chunked GEMMs see dense rows (SURVEY §2.8 long-context strategy).
"""

from __future__ import annotations

from typing import Dict

import torch

__all__ = ["pack_padded_batch", "unpack_values"]


def pack_padded_batch(
    input_ids: torch.Tensor, attention_mask: torch.Tensor
) -> Dict[str, torch.Tensor]:
    """(B, T) padded -> varlen pack.

    Returns dict with:
      packed_ids (1, N) real tokens concatenated,
      position_ids (1, N) restarting at 0 per sequence,
      cu_seqlens (B+1,) cumulative lengths,
      indices (N,) flat gather indices back into the (B*T) layout.
    """
    B, T = input_ids.shape
    mask = attention_mask.bool()
    lengths = mask.sum(dim=1)
    indices = torch.nonzero(mask.reshape(-1), as_tuple=False).squeeze(1)
    packed = input_ids.reshape(-1)[indices].unsqueeze(0)
    cu = torch.zeros(B + 1, dtype=torch.long, device=input_ids.device)
    cu[1:] = torch.cumsum(lengths, dim=0)
    pos = torch.arange(packed.shape[1], device=input_ids.device)
    seq_start = cu[:-1].repeat_interleave(lengths)
    position_ids = (pos - seq_start).unsqueeze(0)
    return {
        "packed_ids": packed,
        "position_ids": position_ids,
        "cu_seqlens": cu,
        "indices": indices,
        "shape": (B, T),
    }


def unpack_values(values: torch.Tensor, pack: Dict[str, torch.Tensor], fill: float = 0.0) -> torch.Tensor:
    """(N,) packed per-token values -> (B, T) padded layout."""
    B, T = pack["shape"]
    out = torch.full((B * T,), fill, dtype=values.dtype, device=values.device)
    out[pack["indices"]] = values.reshape(-1)
    return out.reshape(B, T)
