"""Paged-attention decode: HIP kernel dispatch + eager reference.

The flash-decoding kernel (csrc/paged_attn.hip) reads K/V straight from
the page pool through the page tables (no gather); the eager reference
below is the CPU implementation and the numerics oracle the GPU tests
pin the kernel against.
"""

from __future__ import annotations

import math

import torch

from .backend import extension, use_hip

__all__ = ["paged_attention_decode"]


def _eager_reference(q, k_pool, v_pool, page_table, lengths, scale):
    B, Hq, D = q.shape
    S, Hkv = k_pool.shape[1], k_pool.shape[2]
    gqa = Hq // Hkv
    out = torch.empty(B, Hq, D, dtype=torch.float32, device=q.device)
    qf = q.float()
    for b in range(B):
        n = int(lengths[b])
        pages = page_table[b, : (n + S - 1) // S].long()
        k = k_pool[pages].reshape(-1, Hkv, D)[:n].float()  # (n, Hkv, D)
        v = v_pool[pages].reshape(-1, Hkv, D)[:n].float()
        for hk in range(Hkv):
            for g in range(gqa):
                h = hk * gqa + g
                scores = (k[:, hk] @ qf[b, h]) * scale  # (n,)
                p = torch.softmax(scores, dim=0)
                out[b, h] = p @ v[:, hk]
    return out


def paged_attention_decode(
    q: torch.Tensor,
    k_pool: torch.Tensor,
    v_pool: torch.Tensor,
    page_table: torch.Tensor,
    lengths: torch.Tensor,
    scale: float = None,
    max_len_hint: int = 0,
) -> torch.Tensor:
    """Single-token decode attention over paged K/V.

    q: (B, Hq, D); k_pool/v_pool: (num_pages, page_size, Hkv, D);
    page_table: (B, max_pages) int; lengths: (B,) int.
    ``max_len_hint``: host-side upper bound on the batch's sequence
    lengths; sizes the flash-decoding split count (lengths are device-
    resident, and the table width badly overestimates short decodes).
    Returns (B, Hq, D) fp32.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    ext = extension()
    if use_hip(q) and ext is not None and hasattr(ext, "paged_attn_decode") \
            and q.dtype == torch.bfloat16:
        return ext.paged_attn_decode(q, k_pool, v_pool, page_table, lengths,
                                     float(scale), int(max_len_hint))
    return _eager_reference(q, k_pool, v_pool, page_table, lengths, scale)
