"""Paged single-token decode forward for Llama-family models.

Replaces the decode engine's gather path (pool -> contiguous K/V ->
``DynamicCache`` -> HF forward) with a first-party per-layer step that
reads K/V straight from the paged pool through the page tables via the
``paged_attn_decode`` HIP kernel (ops/csrc/paged_attn.hip) — the role
vLLM's PagedAttention plays in the reference's colocated generation
(``agilerl/algorithms/core/base.py:5421``).

Only the attention data path changes: embeddings, norms, projections
(including any LoRA wrapping — adapter selection is honored because the
projection *modules* are called), MLP and lm_head are the model's own
modules, so logits match the HF forward bit-for-path.  Works on CPU too
(eager reference attention) which is how the parity tests pin it.
"""

from __future__ import annotations

from typing import List

import torch

from ..ops.paged_attn import paged_attention_decode

__all__ = ["PagedLlamaDecodeRunner", "is_paged_decodable"]


def is_paged_decodable(model) -> bool:
    """Structural check: HF Llama-family decoder with separate q/k/v/o
    projections, rotary embedding at the model level."""
    base = getattr(model, "model", None)
    if base is None or not hasattr(base, "layers") or not hasattr(base, "rotary_emb"):
        return False
    try:
        layer = base.layers[0]
    except (IndexError, TypeError):
        return False
    attn = getattr(layer, "self_attn", None)
    return all(
        hasattr(attn, p) for p in ("q_proj", "k_proj", "v_proj", "o_proj")
    ) and hasattr(layer, "input_layernorm") and hasattr(layer, "mlp")


class PagedLlamaDecodeRunner:
    def __init__(self, model, cache):
        self.model = model
        self.cache = cache
        cfg = model.config
        self.n_heads = cfg.num_attention_heads
        self.n_kv = getattr(cfg, "num_key_value_heads", self.n_heads)
        self.head_dim = getattr(cfg, "head_dim", None) or cfg.hidden_size // self.n_heads
        attn0 = model.model.layers[0].self_attn
        self.scale = getattr(attn0, "scaling", self.head_dim ** -0.5)
        self.log2S = (cache.page_size - 1).bit_length()
        assert (1 << self.log2S) == cache.page_size, "page_size must be pow2"

    @torch.no_grad()
    def decode_step(
        self,
        tokens: torch.Tensor,      # (B,) last emitted token per sequence
        positions: torch.Tensor,   # (B,) position of the NEW token (== length)
        table: torch.Tensor,       # (B, max_pages) int32 page table
    ) -> torch.Tensor:             # (B, V) next-token logits
        from transformers.models.llama.modeling_llama import apply_rotary_pos_emb

        model = self.model
        base = model.model
        cache = self.cache
        B = tokens.shape[0]
        Hq, Hkv, D = self.n_heads, self.n_kv, self.head_dim
        S = cache.page_size
        P = cache.num_pages

        x = base.embed_tokens(tokens).unsqueeze(1)               # (B, 1, H)
        cos, sin = base.rotary_emb(x, positions.unsqueeze(1))    # (B, 1, D)

        # flat pool slot of the NEW token for every sequence
        page_of_new = table.gather(
            1, (positions >> self.log2S).unsqueeze(1).long()
        ).squeeze(1).long()
        slots = page_of_new * S + (positions & (S - 1))          # (B,)
        lengths = (positions + 1).to(torch.int32)
        pool_dtype = cache.k_pool.dtype

        for li, layer in enumerate(base.layers):
            h = layer.input_layernorm(x)
            attn = layer.self_attn
            q = attn.q_proj(h).view(B, Hq, D)
            k = attn.k_proj(h).view(B, Hkv, D)
            v = attn.v_proj(h).view(B, Hkv, D)
            qr, kr = apply_rotary_pos_emb(
                q.unsqueeze(2), k.unsqueeze(2), cos, sin, unsqueeze_dim=1
            )
            q = qr.squeeze(2)
            k = kr.squeeze(2)
            # append the new token's K/V into the pool pages in place
            cache.k_pool[li].view(P * S, Hkv, D)[slots] = k.to(pool_dtype)
            cache.v_pool[li].view(P * S, Hkv, D)[slots] = v.to(pool_dtype)
            attn_out = paged_attention_decode(
                q.to(pool_dtype) if pool_dtype == torch.bfloat16 else q,
                cache.k_pool[li], cache.v_pool[li], table, lengths, self.scale,
            )
            attn_out = attn_out.to(x.dtype).view(B, 1, Hq * D)
            x = x + attn.o_proj(attn_out)
            x = x + layer.mlp(layer.post_attention_layernorm(x))
        x = base.norm(x)
        return model.lm_head(x[:, -1])

    # ------------------------------------------------------------------
    def build_table(self, seq_ids: List[int], device: str) -> torch.Tensor:
        """Right-pad the active sequences' page tables into one int32 (B, Pmax)."""
        cache = self.cache
        maxp = max(len(cache.page_tables[s]) for s in seq_ids)
        table = torch.zeros((len(seq_ids), maxp), dtype=torch.int32)
        for i, s in enumerate(seq_ids):
            row = cache.page_tables[s]
            table[i, : len(row)] = torch.tensor(row, dtype=torch.int32)
        return table.to(device)
