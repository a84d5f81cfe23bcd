"""Continuous-batching decode engine over the paged KV cache.

Sequences are admitted the moment pages free up, KV lives in fixed-size
pages behind per-sequence page tables, and finished sequences retire
individually.  On GPU the decode step is fully first-party: a per-layer
forward writes K/V straight into the pool and attends through the
flash-decoding ``paged_attn`` HIP kernel (no gather, no DynamicCache),
the whole step replays as a hipGraph per adapter over static slot
buffers (``llm/paged_llama.py``), token selection is batched (one device
op + one host sync per step), and every generated token's behavior-
policy logprob is captured (the vLLM sampling-logprob analog).  The
eager gather + ``DynamicCache`` path remains as the CPU/fallback
implementation and the parity reference.

Parity contracts (tested): greedy decode through the engine matches
``model.generate(do_sample=False)`` sequence-for-sequence including
ragged admission; graphed and ungraphed engines produce bitwise-equal
sequences under slot churn.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

import torch

from .paged_cache import PagedKVCache

__all__ = ["DecodeEngine", "SequenceState"]


@dataclass
class SequenceState:
    seq_id: int
    prompt_ids: torch.Tensor  # (T,)
    max_new_tokens: int
    temperature: float = 0.0  # 0 => greedy
    adapter: Optional[str] = None
    generated: List[int] = field(default_factory=list)
    # behavior-policy logprob of each generated token (under the sampling
    # temperature) — the decode-engine analog of vLLM sampling logprobs
    gen_logps: List[float] = field(default_factory=list)
    prefilled: bool = False
    done: bool = False

    @property
    def length(self) -> int:
        return self.prompt_ids.numel() + len(self.generated)

    @property
    def last_token(self) -> int:
        return self.generated[-1] if self.generated else int(self.prompt_ids[-1])

    def output_ids(self) -> torch.Tensor:
        gen = torch.tensor(self.generated, dtype=torch.long, device=self.prompt_ids.device)
        return torch.cat([self.prompt_ids, gen])


class DecodeEngine:
    """Batch-of-one-token step loop with continuous admission.

    ``submit()`` any time; ``step()`` prefills newly admitted sequences
    and decodes one token for every active one; finished sequences are
    returned and their pages freed.  ``max_batch`` bounds concurrent
    sequences; excess submissions queue.
    """

    def __init__(
        self,
        model,
        max_batch: int = 64,
        num_pages: int = 512,
        page_size: int = 16,
        eos_token_id: Optional[int] = None,
        set_adapter_fn: Optional[Callable[[Optional[str]], None]] = None,
        use_paged_attention: Optional[bool] = None,
    ):
        self.model = model
        cfg = model.config
        n_kv = getattr(cfg, "num_key_value_heads", cfg.num_attention_heads)
        head_dim = getattr(cfg, "head_dim", cfg.hidden_size // cfg.num_attention_heads)
        dtype = next(model.parameters()).dtype
        device = str(next(model.parameters()).device)
        self.cache = PagedKVCache(
            cfg.num_hidden_layers, n_kv, head_dim,
            num_pages=num_pages, page_size=page_size, dtype=dtype, device=device,
        )
        self.device = device
        # paged-attention decode path: per-layer step reading K/V straight
        # from the pool via the HIP kernel (auto: on for Llama-family on
        # GPU; CPU keeps the gather path unless explicitly requested)
        from .paged_llama import PagedLlamaDecodeRunner, is_paged_decodable

        if use_paged_attention is None:
            use_paged_attention = device.startswith("cuda") and is_paged_decodable(model)
        self._paged_runner = (
            PagedLlamaDecodeRunner(model, self.cache)
            if use_paged_attention and is_paged_decodable(model)
            else None
        )
        # hipGraph-captured decode (GPU): one graph per adapter over static
        # slot buffers; a reserved scratch page absorbs zombie-row writes
        self._graph_decoders: Dict[Optional[str], object] = {}
        self._scratch_page: Optional[int] = None
        self._use_decode_graph = (
            self._paged_runner is not None and device.startswith("cuda")
            and torch.cuda.is_available()
        )
        if self._use_decode_graph and self.cache._free:
            self._scratch_page = self.cache._free.pop()
        self.max_batch = max_batch
        self.eos_token_id = eos_token_id
        self.set_adapter_fn = set_adapter_fn
        self._next_id = 0
        self._reserved: Dict[int, int] = {}  # seq_id -> pages promised
        self.active: Dict[int, SequenceState] = {}
        self.waiting: List[SequenceState] = []
        self.finished_logps: Dict[int, List[float]] = {}  # sampling logprobs
        self.steps_run = 0
        self.tokens_generated = 0
        self.sequences_finished = 0
        self._t_start = None

    # ------------------------------------------------------------------
    def submit(
        self,
        prompt_ids,
        max_new_tokens: int = 32,
        temperature: float = 0.0,
        adapter: Optional[str] = None,
    ) -> int:
        ids = torch.as_tensor(prompt_ids, dtype=torch.long, device=self.device).reshape(-1)
        seq = SequenceState(self._next_id, ids, max_new_tokens, temperature, adapter)
        self._next_id += 1
        self.waiting.append(seq)
        return seq.seq_id

    @property
    def num_active(self) -> int:
        return len(self.active)

    def has_work(self) -> bool:
        return bool(self.active or self.waiting)

    # ------------------------------------------------------------------
    def _available_pages(self) -> int:
        outstanding = sum(
            max(0, need - len(self.cache.page_tables.get(sid, ())))
            for sid, need in self._reserved.items()
        )
        return self.cache.free_pages - outstanding

    def _admit(self) -> None:
        while self.waiting and len(self.active) < self.max_batch:
            seq = self.waiting[0]
            need = self.cache.pages_for(seq.prompt_ids.numel() + seq.max_new_tokens)
            if need > self._available_pages():
                break  # admit in order; wait for pages
            self.waiting.pop(0)
            self.cache.alloc(seq.seq_id)
            self._reserved[seq.seq_id] = need
            self.active[seq.seq_id] = seq

    def _set_adapter(self, adapter: Optional[str]) -> None:
        if self.set_adapter_fn is not None:
            self.set_adapter_fn(adapter)

    @torch.no_grad()
    def _prefill(self, seqs: List[SequenceState]) -> None:
        """Forward full prompts (left-padded batch), store KV pages, and
        emit each sequence's first generated token."""
        from transformers.cache_utils import DynamicCache

        L = max(s.prompt_ids.numel() for s in seqs)
        B = len(seqs)
        ids = torch.zeros((B, L), dtype=torch.long, device=self.device)
        mask = torch.zeros((B, L), dtype=torch.long, device=self.device)
        for i, s in enumerate(seqs):
            n = s.prompt_ids.numel()
            ids[i, L - n :] = s.prompt_ids
            mask[i, L - n :] = 1
        pos = (mask.cumsum(-1) - 1).clamp(min=0)
        cache = DynamicCache()
        out = self.model(
            input_ids=ids, attention_mask=mask, position_ids=pos,
            past_key_values=cache, use_cache=True,
        )
        sel_t = self._select_batch(out.logits[:, -1], seqs)
        lps = self._chosen_logps(out.logits[:, -1], sel_t, seqs).tolist()
        sel = sel_t.tolist()
        # batched pool write: (L, B, n_kv, T, D) stacked once, four tensor
        # ops per layer (vs a python append per sequence per layer)
        k_all = torch.stack([
            self._layer_kv(out.past_key_values, layer)[0]
            for layer in range(self.cache.num_layers)
        ])
        v_all = torch.stack([
            self._layer_kv(out.past_key_values, layer)[1]
            for layer in range(self.cache.num_layers)
        ])
        self.cache.append_prefill([s.seq_id for s in seqs], k_all, v_all, mask)
        for i, s in enumerate(seqs):
            s.prefilled = True
            s.generated.append(int(sel[i]))
            s.gen_logps.append(float(lps[i]))

    @staticmethod
    def _layer_kv(cache, layer: int) -> Tuple[torch.Tensor, torch.Tensor]:
        lay = cache.layers[layer]
        return lay.keys, lay.values

    def _select(self, logits: torch.Tensor, seq: SequenceState) -> int:
        if seq.temperature and seq.temperature > 0:
            probs = torch.softmax(logits / seq.temperature, dim=-1)
            return int(torch.multinomial(probs, 1))
        return int(logits.argmax())

    def _select_batch(self, logits: torch.Tensor, seqs: List[SequenceState]) -> torch.Tensor:
        """One device-side selection for the whole batch (a single host
        sync at .tolist() time instead of one per sequence)."""
        temps = [float(s.temperature or 0.0) for s in seqs]
        greedy = logits.argmax(-1)
        if not any(t > 0 for t in temps):
            return greedy
        t = torch.tensor(
            [tt if tt > 0 else 1.0 for tt in temps], device=logits.device
        )
        probs = torch.softmax(logits.float() / t.unsqueeze(1), dim=-1)
        sampled = torch.multinomial(probs, 1).squeeze(1)
        use = torch.tensor([tt > 0 for tt in temps], device=logits.device)
        return torch.where(use, sampled, greedy)

    @staticmethod
    def _chosen_logps(logits: torch.Tensor, chosen: torch.Tensor,
                      seqs: List[SequenceState]) -> torch.Tensor:
        """Behavior-policy logprob of the chosen tokens (temperature-scaled
        for sampling rows, plain softmax for greedy rows)."""
        t = torch.tensor(
            [float(s.temperature) if s.temperature and s.temperature > 0 else 1.0
             for s in seqs],
            device=logits.device,
        )
        lsm = torch.log_softmax(logits.float() / t.unsqueeze(1), dim=-1)
        return lsm.gather(1, chosen.unsqueeze(1)).squeeze(1)

    @torch.no_grad()
    def _decode_paged(self, seqs: List[SequenceState], adapter=None) -> None:
        """One token for every sequence through the paged-attention path:
        no gather — Q/K/V projections for the single new token, K/V
        appended to the pool in place, attention reads pages directly.
        On GPU the whole step replays a captured hipGraph."""
        runner = self._paged_runner
        seq_ids = [s.seq_id for s in seqs]
        for sid in seq_ids:  # page for the incoming token
            self.cache._ensure_capacity(sid, 1)

        dec = self._graph_decoder(adapter, len(seqs))
        if dec is not None:
            # split sizing: the longest length any active sequence can reach
            hint = max(
                self.cache.lengths[sid] + s.max_new_tokens - len(s.generated)
                for s, sid in zip(seqs, seq_ids)
            )
            if hint > dec.len_hint:
                dec.len_hint = int(hint)
                dec.graph = None  # recapture with the wider split config
            for s, sid in zip(seqs, seq_ids):
                if sid not in dec.slot_of:
                    slot = dec.acquire_slot(sid)
                    dec.load_row(slot, s.last_token, self.cache.lengths[sid],
                                 self.cache.page_tables[sid])
                else:
                    dec.sync_row_pages(dec.slot_of[sid], self.cache.page_tables[sid])
            all_logits = dec.replay()
            slot_t = torch.tensor([dec.slot_of[sid] for sid in seq_ids],
                                  dtype=torch.long, device=self.device)
            row_logits = all_logits[slot_t]
            sel = self._select_batch(row_logits, seqs)
            dec.tokens[slot_t] = sel
            dec.positions[slot_t] += 1
            lps_host = self._chosen_logps(row_logits, sel, seqs).tolist()
            sel_host = sel.tolist()
        else:
            table = runner.build_table(seq_ids, self.device)
            tokens = torch.tensor([s.last_token for s in seqs], dtype=torch.long,
                                  device=self.device)
            positions = torch.tensor([self.cache.lengths[sid] for sid in seq_ids],
                                     dtype=torch.long, device=self.device)
            hint = int(max(self.cache.lengths[sid] for sid in seq_ids)) + 1
            logits = runner.decode_step(tokens, positions, table, max_len_hint=hint)
            sel = self._select_batch(logits, seqs)
            lps_host = self._chosen_logps(logits, sel, seqs).tolist()
            sel_host = sel.tolist()
        for s, sid, tok, lp in zip(seqs, seq_ids, sel_host, lps_host):
            self.cache.lengths[sid] += 1
            s.generated.append(int(tok))
            s.gen_logps.append(float(lp))

    def _graph_decoder(self, adapter, batch_size: int):
        """Per-adapter graphed decoder (adapter routing is python control
        flow inside LoraLinear, so it bakes at capture time)."""
        if not self._use_decode_graph or self._scratch_page is None:
            return None
        from .paged_llama import GraphedPagedDecoder

        dec = self._graph_decoders.get(adapter)
        needed = max(self.max_batch, batch_size)
        if dec is not None and dec.max_batch < needed:
            dec = None  # batch outgrew the captured statics: recapture
        if dec is None:
            dec = GraphedPagedDecoder(
                self._paged_runner, needed, self._scratch_page, self.device
            )
            self._graph_decoders[adapter] = dec
        return dec

    @torch.no_grad()
    def _decode(self, seqs: List[SequenceState], adapter=None) -> None:
        """One token for every sequence: gather pages -> DynamicCache ->
        single batched forward at the shared end position."""
        if self._paged_runner is not None:
            return self._decode_paged(seqs, adapter)
        from transformers.cache_utils import DynamicCache

        seq_ids = [s.seq_id for s in seqs]
        k, v, mask = self.cache.gather(seq_ids)
        legacy = tuple((k[l], v[l]) for l in range(self.cache.num_layers))
        cache = DynamicCache(config=self.model.config)
        for l, (kl, vl) in enumerate(legacy):
            cache.update(kl, vl, l)
        ids = torch.tensor([[s.last_token] for s in seqs], dtype=torch.long, device=self.device)
        full_mask = torch.cat(
            [mask, torch.ones((len(seqs), 1), dtype=torch.long, device=self.device)], dim=1
        )
        pos = torch.tensor([[self.cache.lengths[s.seq_id]] for s in seqs],
                           dtype=torch.long, device=self.device)
        out = self.model(
            input_ids=ids, attention_mask=full_mask, position_ids=pos,
            past_key_values=cache, use_cache=True,
        )
        sel_t = self._select_batch(out.logits[:, -1], seqs)
        lps = self._chosen_logps(out.logits[:, -1], sel_t, seqs).tolist()
        sel = sel_t.tolist()
        for i, s in enumerate(seqs):
            layers_k, layers_v = [], []
            for layer in range(self.cache.num_layers):
                k_l, v_l = self._layer_kv(out.past_key_values, layer)
                layers_k.append(k_l[i, :, -1:].permute(1, 0, 2))
                layers_v.append(v_l[i, :, -1:].permute(1, 0, 2))
            self.cache.append(s.seq_id, torch.stack(layers_k), torch.stack(layers_v))
            s.generated.append(int(sel[i]))
            s.gen_logps.append(float(lps[i]))

    # ------------------------------------------------------------------
    def step(self) -> List[Tuple[int, torch.Tensor]]:
        """Admit, prefill, decode one token each, retire finished.

        Returns [(seq_id, full_ids)] for sequences that finished this step.
        """
        import time as _time

        self._admit()
        if not self.active:
            return []
        if self._t_start is None:
            self._t_start = _time.time()
        self.steps_run += 1
        by_adapter: Dict[Optional[str], List[SequenceState]] = {}
        for s in self.active.values():
            by_adapter.setdefault(s.adapter, []).append(s)
        for adapter, seqs in by_adapter.items():
            self._set_adapter(adapter)
            new = [s for s in seqs if not s.prefilled]
            old = [s for s in seqs if s.prefilled]
            if new:
                # the token the last prefill/decode emitted has no KV yet;
                # it becomes this step's decode input for old sequences
                self._prefill(new)
            if old:
                self._decode(old, adapter)
        finished = []
        for s in list(self.active.values()):
            hit_eos = self.eos_token_id is not None and s.generated and \
                s.generated[-1] == self.eos_token_id
            if len(s.generated) >= s.max_new_tokens or hit_eos:
                s.done = True
                self.sequences_finished += 1
                self.tokens_generated += len(s.generated)
                finished.append((s.seq_id, s.output_ids()))
                self.finished_logps[s.seq_id] = list(s.gen_logps)
                self._release_seq(s.seq_id)
        return finished

    def _release_seq(self, seq_id: int) -> None:
        self.cache.free(seq_id)
        self._reserved.pop(seq_id, None)
        self.active.pop(seq_id, None)
        for dec in self._graph_decoders.values():
            dec.release_slot(seq_id)

    def cancel(self, seq_id: int) -> bool:
        """Abort a sequence (waiting or active); frees its pages."""
        for i, w in enumerate(self.waiting):
            if w.seq_id == seq_id:
                self.waiting.pop(i)
                return True
        if seq_id in self.active:
            self._release_seq(seq_id)
            return True
        return False

    def stats(self) -> Dict[str, float]:
        """Decode observability: steps, finished sequences, tokens/s."""
        import time as _time

        elapsed = (_time.time() - self._t_start) if self._t_start else 0.0
        return {
            "steps": self.steps_run,
            "active": len(self.active),
            "waiting": len(self.waiting),
            "sequences_finished": self.sequences_finished,
            "tokens_generated": self.tokens_generated,
            "tokens_per_sec": self.tokens_generated / elapsed if elapsed > 0 else 0.0,
            "free_pages": self.cache.free_pages,
        }

    def run_all(self, max_steps: int = 10_000) -> Dict[int, torch.Tensor]:
        """Drive until every submitted sequence finishes."""
        results: Dict[int, torch.Tensor] = {}
        for _ in range(max_steps):
            if not self.has_work():
                break
            for seq_id, ids in self.step():
                results[seq_id] = ids
        return results
