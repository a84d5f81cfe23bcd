"""Evolvable GPT (decoder-only transformer).

Reference parity: ``agilerl/modules/gpt.py:22`` (EvolvableGPT with
``CausalSelfAttention`` :719; used by the legacy ILQL / BC_LM offline
stack).  Attention runs through ``F.scaled_dot_product_attention`` (the
ROCm SDPA/flash path on MI355X); depth and width evolve with parameter
preservation.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from .base import EvolvableModule, MutationType, mutation, preserve_parameters
from .components import NewGELU

__all__ = ["EvolvableGPT", "CausalSelfAttention", "GPTBlock"]


class CausalSelfAttention(nn.Module):
    def __init__(self, n_embd: int, n_head: int, dropout: float = 0.0):
        super().__init__()
        assert n_embd % n_head == 0
        self.n_head = n_head
        self.qkv = nn.Linear(n_embd, 3 * n_embd)
        self.proj = nn.Linear(n_embd, n_embd)
        self.dropout = dropout

    def forward(self, x: torch.Tensor, is_causal: bool = True) -> torch.Tensor:
        B, T, C = x.shape
        q, k, v = self.qkv(x).split(C, dim=2)
        q = q.view(B, T, self.n_head, C // self.n_head).transpose(1, 2)
        k = k.view(B, T, self.n_head, C // self.n_head).transpose(1, 2)
        v = v.view(B, T, self.n_head, C // self.n_head).transpose(1, 2)
        y = F.scaled_dot_product_attention(
            q, k, v, dropout_p=self.dropout if self.training else 0.0, is_causal=is_causal
        )
        y = y.transpose(1, 2).contiguous().view(B, T, C)
        return self.proj(y)


class GPTBlock(nn.Module):
    def __init__(self, n_embd: int, n_head: int, dropout: float = 0.0, causal: bool = True,
                 dim_feedfwd: int = 0, activation: str = "NewGELU",
                 layer_norm_eps: float = 1e-5, bias: bool = True):
        super().__init__()
        from .components import get_activation

        ffwd = dim_feedfwd or 4 * n_embd
        self.ln1 = nn.LayerNorm(n_embd, eps=layer_norm_eps, bias=bias)
        self.attn = CausalSelfAttention(n_embd, n_head, dropout)
        self.ln2 = nn.LayerNorm(n_embd, eps=layer_norm_eps, bias=bias)
        self.mlp = nn.Sequential(
            nn.Linear(n_embd, ffwd, bias=bias), get_activation(activation),
            nn.Linear(ffwd, n_embd, bias=bias), nn.Dropout(dropout),
        )
        self.causal = causal

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.attn(self.ln1(x), is_causal=self.causal)
        x = x + self.mlp(self.ln2(x))
        return x


class EvolvableGPT(EvolvableModule):
    CAUSAL = True

    def __init__(
        self,
        vocab_size: int,
        n_layer: int = 4,
        n_head: int = 4,
        n_embd: int = 128,
        max_positions: int = 512,
        block_size: Optional[int] = None,
        dropout: float = 0.0,
        dim_feedfwd: int = 0,
        activation: str = "NewGELU",
        layer_norm_eps: float = 1e-5,
        bias: bool = True,
        min_layers: int = 1,
        max_layers: int = 12,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        # reference modules/gpt.py spells the context window `block_size`
        if block_size is not None:
            max_positions = int(block_size)
        self.dim_feedfwd = int(dim_feedfwd)
        self.activation = activation
        self.layer_norm_eps = float(layer_norm_eps)
        self.bias = bool(bias)
        self.vocab_size = int(vocab_size)
        self.n_layer = int(n_layer)
        self.n_head = int(n_head)
        self.n_embd = int(n_embd)
        self.max_positions = int(max_positions)
        self.dropout = dropout
        self.min_layers = min_layers
        self.max_layers = max_layers
        self.model = self._build().to(device)

    def _build(self) -> nn.ModuleDict:
        return nn.ModuleDict(
            dict(
                wte=nn.Embedding(self.vocab_size, self.n_embd),
                wpe=nn.Embedding(self.max_positions, self.n_embd),
                blocks=nn.ModuleList(
                    GPTBlock(self.n_embd, self.n_head, self.dropout, self.CAUSAL,
                             dim_feedfwd=self.dim_feedfwd, activation=self.activation,
                             layer_norm_eps=self.layer_norm_eps, bias=self.bias)
                    for _ in range(self.n_layer)
                ),
                ln_f=nn.LayerNorm(self.n_embd, eps=self.layer_norm_eps, bias=self.bias),
                head=nn.Linear(self.n_embd, self.vocab_size, bias=False),
            )
        )

    def transformer_forward(self, idx: torch.Tensor) -> torch.Tensor:
        """(B, T) token ids -> (B, T, n_embd) hidden states."""
        B, T = idx.shape
        pos = torch.arange(T, device=idx.device)
        x = self.model["wte"](idx) + self.model["wpe"](pos).unsqueeze(0)
        for block in self.model["blocks"]:
            x = block(x)
        return self.model["ln_f"](x)

    def forward(
        self, idx: torch.Tensor, targets: Optional[torch.Tensor] = None
    ):
        hidden = self.transformer_forward(idx)
        logits = self.model["head"](hidden)
        if targets is None:
            return logits
        loss = F.cross_entropy(
            logits.reshape(-1, self.vocab_size), targets.reshape(-1), ignore_index=-1
        )
        return logits, loss

    @torch.no_grad()
    def generate(self, idx: torch.Tensor, max_new_tokens: int, temperature: float = 1.0,
                 top_k: Optional[int] = None) -> torch.Tensor:
        for _ in range(max_new_tokens):
            ctx = idx[:, -self.max_positions :]
            logits = self(ctx)[:, -1, :] / max(temperature, 1e-6)
            if top_k is not None:
                v, _ = torch.topk(logits, min(top_k, logits.size(-1)))
                logits[logits < v[:, [-1]]] = -float("inf")
            probs = F.softmax(logits, dim=-1)
            idx = torch.cat([idx, torch.multinomial(probs, 1)], dim=1)
        return idx

    @property
    def output_size(self) -> int:
        return self.vocab_size

    def reset_noise(self) -> None:
        pass

    # ------------------------------------------------------------------
    def recreate_network(self) -> None:
        new_model = self._build().to(self.device)
        preserve_parameters(self.model, new_model)
        self.model = new_model

    @mutation(MutationType.LAYER)
    def add_layer(self) -> dict:
        if self.n_layer < self.max_layers:
            self.n_layer += 1
            self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def remove_layer(self) -> dict:
        if self.n_layer > self.min_layers:
            self.n_layer -= 1
            self.recreate_network()
        return {}

    @mutation(MutationType.NODE)
    def add_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([32, 64])) * 1
        # widen in head-size multiples so n_embd stays divisible by n_head
        numb_new_nodes = max(numb_new_nodes // self.n_head, 1) * self.n_head
        if self.n_embd + numb_new_nodes <= 1024:
            self.n_embd += numb_new_nodes
            self.recreate_network()
        return {"numb_new_nodes": numb_new_nodes}

    @mutation(MutationType.NODE)
    def remove_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([32, 64]))
        numb_new_nodes = max(numb_new_nodes // self.n_head, 1) * self.n_head
        if self.n_embd - numb_new_nodes >= 32:
            self.n_embd -= numb_new_nodes
            self.recreate_network()
        return {"numb_new_nodes": numb_new_nodes}
