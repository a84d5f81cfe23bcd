"""Evolvable BERT: encoder-decoder transformer (encoder-only mode kept).

Reference parity: ``agilerl/modules/bert.py:63`` (EvolvableBERT) — an
end-to-end seq2seq transformer: per-layer feedforward widths
(``encoder_layers``/``decoder_layers`` lists), separate source/target
vocabularies, optional final norms, layer add/remove mutations on both
stacks.  ``decoder_layers=None`` (with the legacy ``vocab_size``/
``n_layer``/``n_embd`` spellings) keeps the round-1 encoder-only mode
with a pooled [CLS]-style classification head.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .base import EvolvableModule, MutationType, mutation, preserve_parameters
from .components import get_activation
from .gpt import CausalSelfAttention, GPTBlock

__all__ = ["EvolvableBERT"]


class CrossAttention(nn.Module):
    """Decoder->memory attention (queries from tgt, keys/values from src)."""

    def __init__(self, d_model: int, n_head: int, dropout: float = 0.0):
        super().__init__()
        assert d_model % n_head == 0
        self.n_head = n_head
        self.q = nn.Linear(d_model, d_model)
        self.kv = nn.Linear(d_model, 2 * d_model)
        self.proj = nn.Linear(d_model, d_model)
        self.dropout = dropout

    def forward(self, x: torch.Tensor, memory: torch.Tensor) -> torch.Tensor:
        B, T, C = x.shape
        S = memory.shape[1]
        q = self.q(x).view(B, T, self.n_head, C // self.n_head).transpose(1, 2)
        k, v = self.kv(memory).split(C, dim=2)
        k = k.view(B, S, self.n_head, C // self.n_head).transpose(1, 2)
        v = v.view(B, S, self.n_head, C // self.n_head).transpose(1, 2)
        y = F.scaled_dot_product_attention(
            q, k, v, dropout_p=self.dropout if self.training else 0.0
        )
        return self.proj(y.transpose(1, 2).contiguous().view(B, T, C))


class DecoderBlock(nn.Module):
    """Causal self-attn + cross-attn + feedforward (pre-LN)."""

    def __init__(self, d_model: int, n_head: int, ffwd: int, dropout: float,
                 activation: str, layer_norm_eps: float):
        super().__init__()
        self.ln1 = nn.LayerNorm(d_model, eps=layer_norm_eps)
        self.self_attn = CausalSelfAttention(d_model, n_head, dropout)
        self.ln2 = nn.LayerNorm(d_model, eps=layer_norm_eps)
        self.cross_attn = CrossAttention(d_model, n_head, dropout)
        self.ln3 = nn.LayerNorm(d_model, eps=layer_norm_eps)
        self.mlp = nn.Sequential(
            nn.Linear(d_model, ffwd), get_activation(activation),
            nn.Linear(ffwd, d_model), nn.Dropout(dropout),
        )

    def forward(self, x: torch.Tensor, memory: torch.Tensor) -> torch.Tensor:
        x = x + self.self_attn(self.ln1(x), is_causal=True)
        x = x + self.cross_attn(self.ln2(x), memory)
        return x + self.mlp(self.ln3(x))


class EvolvableBERT(EvolvableModule):
    def __init__(
        self,
        encoder_layers: Optional[List[int]] = None,
        decoder_layers: Optional[List[int]] = None,
        end2end: bool = True,
        src_vocab_size: Optional[int] = None,
        tgt_vocab_size: Optional[int] = None,
        encoder_norm: bool = True,
        decoder_norm: bool = True,
        d_model: int = 128,
        n_head: int = 8,
        dropout: float = 0.0,
        max_positions: int = 512,
        num_outputs: Optional[int] = None,
        min_layers: int = 1,
        max_encoder_layers: int = 12,
        max_decoder_layers: int = 12,
        layer_norm_eps: float = 1e-5,
        activation: str = "NewGELU",
        batch_first: bool = True,
        norm_first: bool = False,
        # legacy (round-1 encoder-only) spellings
        vocab_size: Optional[int] = None,
        n_layer: Optional[int] = None,
        n_embd: Optional[int] = None,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        if not batch_first:
            raise ValueError("only batch_first=True layouts are supported")
        if n_embd is not None:
            d_model = int(n_embd)
        if vocab_size is not None:
            src_vocab_size = src_vocab_size or int(vocab_size)
        if src_vocab_size is None:
            raise ValueError("src_vocab_size (or legacy vocab_size) is required")
        if encoder_layers is None:
            encoder_layers = [4 * d_model] * int(n_layer if n_layer is not None else 4)
        self.encoder_layers = [int(f) for f in encoder_layers]
        self.decoder_layers = [int(f) for f in (decoder_layers or [])]
        self.end2end = bool(end2end)
        self.src_vocab_size = int(src_vocab_size)
        self.tgt_vocab_size = int(tgt_vocab_size or src_vocab_size)
        self.encoder_norm = bool(encoder_norm)
        self.decoder_norm = bool(decoder_norm)
        self.d_model = int(d_model)
        self.n_head = int(n_head)
        self.dropout = float(dropout)
        self.max_positions = int(max_positions)
        self.num_outputs = num_outputs
        self.min_layers = int(min_layers)
        self.max_encoder_layers = int(max_encoder_layers)
        self.max_decoder_layers = int(max_decoder_layers)
        self.layer_norm_eps = float(layer_norm_eps)
        self.activation = activation
        self.norm_first = bool(norm_first)
        self.model = self._build().to(device)

    # ------------------------------------------------------------------
    def _build(self) -> nn.ModuleDict:
        parts = dict(
            src_tok=nn.Embedding(self.src_vocab_size, self.d_model),
            src_pos=nn.Embedding(self.max_positions, self.d_model),
            encoder=nn.ModuleList(
                GPTBlock(self.d_model, self.n_head, self.dropout, causal=False,
                         dim_feedfwd=f, activation=self.activation,
                         layer_norm_eps=self.layer_norm_eps)
                for f in self.encoder_layers
            ),
        )
        if self.encoder_norm:
            parts["enc_norm"] = nn.LayerNorm(self.d_model, eps=self.layer_norm_eps)
        if self.decoder_layers:
            parts.update(
                tgt_tok=nn.Embedding(self.tgt_vocab_size, self.d_model),
                tgt_pos=nn.Embedding(self.max_positions, self.d_model),
                decoder=nn.ModuleList(
                    DecoderBlock(self.d_model, self.n_head, f, self.dropout,
                                 self.activation, self.layer_norm_eps)
                    for f in self.decoder_layers
                ),
                generator=nn.Linear(self.d_model, self.tgt_vocab_size),
            )
            if self.decoder_norm:
                parts["dec_norm"] = nn.LayerNorm(self.d_model, eps=self.layer_norm_eps)
        elif self.num_outputs is not None:
            parts["cls_head"] = nn.Linear(self.d_model, self.num_outputs)
        else:
            parts["generator"] = nn.Linear(self.d_model, self.src_vocab_size, bias=False)
        return nn.ModuleDict(parts)

    # ------------------------------------------------------------------
    def encode(self, src: torch.Tensor) -> torch.Tensor:
        pos = torch.arange(src.shape[1], device=src.device)
        x = self.model["src_tok"](src) + self.model["src_pos"](pos).unsqueeze(0)
        for block in self.model["encoder"]:
            x = block(x)
        if "enc_norm" in self.model:
            x = self.model["enc_norm"](x)
        return x

    def decode(self, tgt: torch.Tensor, memory: torch.Tensor) -> torch.Tensor:
        pos = torch.arange(tgt.shape[1], device=tgt.device)
        x = self.model["tgt_tok"](tgt) + self.model["tgt_pos"](pos).unsqueeze(0)
        for block in self.model["decoder"]:
            x = block(x, memory)
        if "dec_norm" in self.model:
            x = self.model["dec_norm"](x)
        return x

    def forward(self, src: torch.Tensor, tgt: Optional[torch.Tensor] = None,
                targets: Optional[torch.Tensor] = None):
        memory = self.encode(src)
        if self.decoder_layers:
            if tgt is None:
                raise ValueError("seq2seq EvolvableBERT needs forward(src, tgt)")
            hidden = self.decode(tgt, memory)
            logits = self.model["generator"](hidden)
        elif self.num_outputs is not None and targets is None:
            return self.model["cls_head"](memory[:, 0])  # pooled first token
        else:
            logits = self.model["generator"](memory)
        if targets is not None:
            loss = F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), targets.reshape(-1),
                ignore_index=-1,
            )
            return logits, loss
        return logits

    @property
    def output_size(self) -> int:
        if self.decoder_layers:
            return self.tgt_vocab_size
        return self.num_outputs if self.num_outputs is not None else self.src_vocab_size

    def reset_noise(self) -> None:
        pass

    # ------------------------------------------------------------------
    def recreate_network(self) -> None:
        new_model = self._build().to(self.device)
        for key, mod in new_model.items():
            if key in self.model:
                preserve_parameters(self.model[key], mod)
        self.model = new_model

    @mutation(MutationType.LAYER)
    def add_encoder_layer(self) -> dict:
        if len(self.encoder_layers) < self.max_encoder_layers:
            self.encoder_layers.append(self.encoder_layers[-1])
            self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def remove_encoder_layer(self) -> dict:
        if len(self.encoder_layers) > self.min_layers:
            self.encoder_layers.pop()
            self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def add_decoder_layer(self) -> dict:
        if self.decoder_layers and len(self.decoder_layers) < self.max_decoder_layers:
            self.decoder_layers.append(self.decoder_layers[-1])
            self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def remove_decoder_layer(self) -> dict:
        if len(self.decoder_layers) > self.min_layers:
            self.decoder_layers.pop()
            self.recreate_network()
        return {}

    @mutation(MutationType.NODE)
    def add_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        """Widen every feedforward by ``numb_new_nodes`` (default 32)."""
        n = int(numb_new_nodes or 32)
        self.encoder_layers = [f + n for f in self.encoder_layers]
        self.decoder_layers = [f + n for f in self.decoder_layers]
        self.recreate_network()
        return {"numb_new_nodes": n}
