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
import torch.nn.functional as F

from ..ops.backend import extension, use_hip
from ..ops.paged_attn import paged_attention_decode

__all__ = ["PagedLlamaDecodeRunner", "is_paged_decodable"]


def _fast_linear(module, x2d: torch.Tensor) -> torch.Tensor:
    """Decode-shape projection: route the big weight stream through the
    skinny-GEMM kernel (ops/csrc/skinny_gemm.hip) instead of hipBLASLt's
    tile kernels (~0.6-1 TB/s at M<=16), keeping any LoRA delta as the
    small eager side-path exactly like LoraLinear.forward."""
    ext = extension()
    base = getattr(module, "base", module)  # LoraLinear wraps .base
    eligible = (
        ext is not None
        and hasattr(ext, "skinny_gemm")
        and use_hip(x2d)
        and x2d.dtype == torch.bfloat16
        and x2d.shape[0] <= 16
        and x2d.shape[1] % 8 == 0
        # the kernel beats hipBLASLt only below its small-GEMM floor
        # (~19 us): N<=2048 covers the GQA k/v projections; larger shapes
        # stream at 5.5+ TB/s on hipBLASLt and stay there
        and getattr(base, "out_features", 1 << 30) <= 2048
    )
    if (
        not eligible
        or not isinstance(base, torch.nn.Linear)
        or base.bias is not None
        or base.weight.dtype != torch.bfloat16
    ):
        return module(x2d)
    out = ext.skinny_gemm(x2d, base.weight)
    name = getattr(module, "active_adapter", None)
    if (
        name is not None
        and name in getattr(module, "lora_A", ())
        and getattr(module, "_merged", None) is None
    ):
        h = module.dropout(x2d) if getattr(module, "dropout", None) is not None else x2d
        out = out + F.linear(F.linear(h, module.lora_A[name]), module.lora_B[name]) * module.scaling
    return out


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
        max_len_hint: int = 0,     # host-side length upper bound (split sizing)
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
            h2 = h.view(B, -1)
            q = _fast_linear(attn.q_proj, h2).view(B, Hq, D)
            k = _fast_linear(attn.k_proj, h2).view(B, Hkv, D)
            v = _fast_linear(attn.v_proj, h2).view(B, Hkv, D)
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
                max_len_hint=max_len_hint,
            )
            attn_out = attn_out.to(x.dtype).view(B, Hq * D)
            x = x + _fast_linear(attn.o_proj, attn_out).view(B, 1, -1)
            x = x + self._mlp(layer.mlp, layer.post_attention_layernorm(x), B)
        x = base.norm(x)
        return _fast_linear(model.lm_head, x[:, -1])

    def _mlp(self, mlp, h: torch.Tensor, B: int) -> torch.Tensor:
        """Llama MLP with skinny-GEMM projections + fused SwiGLU when the
        module has the standard gate/up/down layout."""
        if all(hasattr(mlp, a) for a in ("gate_proj", "up_proj", "down_proj")):
            from ..ops.swiglu import swiglu

            h2 = h.view(B, -1)
            gate = _fast_linear(mlp.gate_proj, h2)
            up = _fast_linear(mlp.up_proj, h2)
            return _fast_linear(mlp.down_proj, swiglu(gate, up)).view(B, 1, -1)
        return mlp(h)

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


class GraphedPagedDecoder:
    """hipGraph-captured decode step over static slot buffers.

    The per-layer python loop costs ~10 module calls x n_layers of launch
    dispatch per decode step (~320 launches for an 8B model) — far more
    than the ~2 ms of actual HBM work.  This captures ONE whole
    ``decode_step`` into a hipGraph over static (tokens, positions, table)
    buffers sized to ``max_batch`` rows, so a decode step becomes: update
    slot buffers in place -> replay -> batched token selection.

    Inactive ("zombie") slots point at a reserved scratch page with
    length 1: replays do a little wasted work on them instead of forcing
    a recapture whenever the active set changes.  Admission just writes
    the new row's slot buffers; recapture only happens if the engine is
    rebuilt (e.g. after evolution clones) or the adapter set changes
    (adapter routing is python control flow, baked at capture).
    """

    def __init__(self, runner: PagedLlamaDecodeRunner, max_batch: int,
                 scratch_page: int, device: str):
        cache = runner.cache
        B = max_batch
        maxp = min(
            cache.num_pages,
            cache.pages_for(getattr(runner.model.config, "max_position_embeddings", 8192)),
        )
        self.runner = runner
        self.device = device
        self.max_batch = B
        self.max_pages = maxp
        self.scratch_page = scratch_page
        self.tokens = torch.zeros(B, dtype=torch.long, device=device)
        self.positions = torch.zeros(B, dtype=torch.long, device=device)
        self.table = torch.full((B, maxp), scratch_page, dtype=torch.int32, device=device)
        self.logits: torch.Tensor = None
        self.graph = None
        self.len_hint = 0              # max expected sequence length (splits)
        self.slot_of = {}              # seq_id -> row
        self._free_slots = list(range(B - 1, -1, -1))
        self._pages_synced = [0] * B   # table entries filled per row

    def acquire_slot(self, seq_id: int) -> int:
        slot = self._free_slots.pop()
        self.slot_of[seq_id] = slot
        return slot

    def release_slot(self, seq_id: int) -> None:
        slot = self.slot_of.pop(seq_id, None)
        if slot is not None:
            self.reset_row(slot)
            self._free_slots.append(slot)

    def reset_row(self, i: int) -> None:
        """Return slot i to zombie state (scratch page, position 0)."""
        self.tokens[i] = 0
        self.positions[i] = 0
        self.table[i].fill_(self.scratch_page)
        self._pages_synced[i] = 0

    def load_row(self, i: int, token: int, position: int, pages: List[int]) -> None:
        self.tokens[i] = token
        self.positions[i] = position
        row = torch.full((self.max_pages,), self.scratch_page, dtype=torch.int32)
        row[: len(pages)] = torch.tensor(pages, dtype=torch.int32)
        self.table[i].copy_(row.to(self.device))
        self._pages_synced[i] = len(pages)

    def sync_row_pages(self, i: int, pages: List[int]) -> None:
        """Reflect newly-allocated pages for slot i into the static table."""
        n = len(pages)
        if n == self._pages_synced[i]:
            return
        new = pages[self._pages_synced[i]:]
        self.table[i, self._pages_synced[i]: n] = torch.tensor(
            new, dtype=torch.int32
        ).to(self.device)
        self._pages_synced[i] = n

    @torch.no_grad()
    def capture(self) -> None:
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):  # warmup allocations on a side stream
                self.runner.decode_step(self.tokens, self.positions, self.table,
                                        max_len_hint=self.len_hint)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.logits = self.runner.decode_step(
                self.tokens, self.positions, self.table, max_len_hint=self.len_hint)
        torch.cuda.synchronize()

    @torch.no_grad()
    def replay(self) -> torch.Tensor:
        if self.graph is None:
            self.capture()
        self.graph.replay()
        return self.logits
