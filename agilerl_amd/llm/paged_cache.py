"""Paged KV cache: fixed-size HBM pages + per-sequence page tables.

Pages live in a preallocated pool; sequences own page lists.  The GPU
decode path reads the pool DIRECTLY through the page tables inside the
``paged_attn`` HIP kernel and appends each new token's K/V in place
(``llm/paged_llama.py``); prefill appends are batched tensor writes
(:meth:`append_prefill`).  :meth:`gather` remains for the CPU/fallback
decode path.  Reference analog: vLLM's PagedAttention block manager
(the reference colocates vLLM for generation; here it is first-party).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

__all__ = ["PagedKVCache"]


class PagedKVCache:
    """One pool per model: (num_layers, num_pages, page_size, n_kv, D)."""

    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        num_pages: int = 256,
        page_size: int = 16,
        dtype: torch.dtype = torch.float32,
        device: str = "cpu",
    ):
        self.num_layers = num_layers
        self.num_kv_heads = num_kv_heads
        self.head_dim = head_dim
        self.num_pages = num_pages
        self.page_size = page_size
        self.device = device
        shape = (num_layers, num_pages, page_size, num_kv_heads, head_dim)
        self.k_pool = torch.zeros(shape, dtype=dtype, device=device)
        self.v_pool = torch.zeros(shape, dtype=dtype, device=device)
        self._free: List[int] = list(range(num_pages - 1, -1, -1))
        self.page_tables: Dict[int, List[int]] = {}
        self.lengths: Dict[int, int] = {}

    # ------------------------------------------------------------------
    @property
    def free_pages(self) -> int:
        return len(self._free)

    def pages_for(self, tokens: int) -> int:
        return (tokens + self.page_size - 1) // self.page_size

    def alloc(self, seq_id: int) -> None:
        if seq_id in self.page_tables:
            raise KeyError(f"sequence {seq_id} already allocated")
        self.page_tables[seq_id] = []
        self.lengths[seq_id] = 0

    def free(self, seq_id: int) -> None:
        self._free.extend(reversed(self.page_tables.pop(seq_id, [])))
        self.lengths.pop(seq_id, None)

    def _ensure_capacity(self, seq_id: int, new_tokens: int) -> None:
        table = self.page_tables[seq_id]
        need = self.pages_for(self.lengths[seq_id] + new_tokens)
        while len(table) < need:
            if not self._free:
                raise RuntimeError("PagedKVCache: out of pages")
            table.append(self._free.pop())

    # ------------------------------------------------------------------
    def append(self, seq_id: int, k: torch.Tensor, v: torch.Tensor) -> None:
        """Append T tokens of K/V for every layer.

        k, v: (num_layers, T, n_kv, D).
        """
        T = k.shape[1]
        self._ensure_capacity(seq_id, T)
        table = self.page_tables[seq_id]
        pos = self.lengths[seq_id]
        done = 0
        while done < T:
            page = table[(pos + done) // self.page_size]
            off = (pos + done) % self.page_size
            n = min(self.page_size - off, T - done)
            self.k_pool[:, page, off : off + n] = k[:, done : done + n]
            self.v_pool[:, page, off : off + n] = v[:, done : done + n]
            done += n
        self.lengths[seq_id] = pos + T

    @torch.no_grad()
    def append_prefill(self, seq_ids, k, v, mask) -> None:
        """Batched first-append for freshly-admitted sequences.

        k/v: (num_layers, B, n_kv, T, D) as produced by the prefill
        forward; mask: (B, T) with 1 on real (right-aligned) positions.
        Four tensor ops per layer instead of a python stack+append per
        sequence (the per-step decode path appends in-kernel already).
        Sequences must have length 0 (just allocated).
        """
        L = self.num_layers
        S = self.page_size
        lengths = mask.sum(dim=1).tolist()
        slot_rows = []
        for sid, n in zip(seq_ids, lengths):
            n = int(n)
            if self.lengths[sid] != 0:
                raise RuntimeError("append_prefill requires fresh sequences")
            self._ensure_capacity(sid, n)
            pos = torch.arange(n)
            pages = torch.tensor(self.page_tables[sid], dtype=torch.long)[
                torch.div(pos, S, rounding_mode="floor")
            ]
            slot_rows.append(pages * S + pos % S)
        slots = torch.cat(slot_rows).to(self.device) if slot_rows else None
        if slots is None or slots.numel() == 0:
            return
        valid = mask.bool()
        flat_k = self.k_pool.view(L, self.num_pages * S, self.num_kv_heads, self.head_dim)
        flat_v = self.v_pool.view(L, self.num_pages * S, self.num_kv_heads, self.head_dim)
        for layer in range(L):
            # (B, n_kv, T, D) -> (B, T, n_kv, D) -> (total, n_kv, D)
            flat_k[layer][slots] = k[layer].permute(0, 2, 1, 3)[valid].to(self.k_pool.dtype)
            flat_v[layer][slots] = v[layer].permute(0, 2, 1, 3)[valid].to(self.v_pool.dtype)
        for sid, n in zip(seq_ids, lengths):
            self.lengths[sid] = int(n)

    def gather(
        self, seq_ids: List[int], pad_to: Optional[int] = None
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """Left-padded contiguous K/V for a batch of sequences.

        Returns (k, v, mask): k/v (num_layers, B, n_kv, pad_to, D);
        mask (B, pad_to) with 1 on real positions (right-aligned).
        """
        B = len(seq_ids)
        L = pad_to or max(self.lengths[s] for s in seq_ids)
        k = self.k_pool.new_zeros((self.num_layers, B, self.num_kv_heads, L, self.head_dim))
        v = torch.zeros_like(k)
        mask = torch.zeros((B, L), dtype=torch.long, device=self.device)
        for i, s in enumerate(seq_ids):
            n = self.lengths[s]
            if n == 0:
                continue
            pages = self.page_tables[s][: self.pages_for(n)]
            kk = self.k_pool[:, pages].reshape(self.num_layers, -1, self.num_kv_heads, self.head_dim)[:, :n]
            vv = self.v_pool[:, pages].reshape(self.num_layers, -1, self.num_kv_heads, self.head_dim)[:, :n]
            k[:, i, :, L - n :] = kk.permute(0, 2, 1, 3)
            v[:, i, :, L - n :] = vv.permute(0, 2, 1, 3)
            mask[i, L - n :] = 1
        return k, v, mask
