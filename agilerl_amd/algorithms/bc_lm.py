"""BC_LM — behavior-cloning language model (legacy offline stack).

Reference parity: ``agilerl/algorithms/bc_lm.py:38`` (BC on EvolvableGPT
with top-k/top-p sampling utilities).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..modules.gpt import EvolvableGPT

__all__ = ["BC_LM"]


class BC_LM(nn.Module):
    def __init__(
        self,
        vocab_size: int,
        n_layer: int = 4,
        n_head: int = 4,
        n_embd: int = 128,
        max_positions: int = 512,
        lr: float = 3e-4,
        device: str = "cpu",
    ):
        super().__init__()
        self.vocab_size = vocab_size
        self.device = device
        self.gpt = EvolvableGPT(
            vocab_size, n_layer=n_layer, n_head=n_head, n_embd=n_embd,
            max_positions=max_positions, device=device,
        )
        self.optimizer = torch.optim.AdamW(self.parameters(), lr=lr)

    def learn(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        ids = batch["ids"].to(self.device)
        mask = batch.get("mask")
        targets = ids[:, 1:].clone()
        if mask is not None:
            targets[mask.to(self.device) < 0.5] = -1  # ignore index
        _, loss = self.gpt(ids[:, :-1], targets)
        self.optimizer.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.parameters(), 1.0)
        self.optimizer.step()
        return {"loss": float(loss.detach())}

    @torch.no_grad()
    def generate(
        self,
        idx: torch.Tensor,
        max_new_tokens: int,
        temperature: float = 1.0,
        top_k: Optional[int] = None,
        top_p: Optional[float] = None,
    ) -> torch.Tensor:
        for _ in range(max_new_tokens):
            ctx = idx[:, -self.gpt.max_positions :]
            logits = self.gpt(ctx)[:, -1, :] / max(temperature, 1e-6)
            if top_k is not None:
                v, _ = torch.topk(logits, min(top_k, logits.size(-1)))
                logits[logits < v[:, [-1]]] = -float("inf")
            if top_p is not None:
                sorted_logits, sorted_idx = torch.sort(logits, descending=True)
                cum = torch.softmax(sorted_logits, dim=-1).cumsum(dim=-1)
                cut = cum > top_p
                cut[:, 1:] = cut[:, :-1].clone()
                cut[:, 0] = False
                remove = cut.scatter(1, sorted_idx, cut)
                logits[remove] = -float("inf")
            probs = F.softmax(logits, dim=-1)
            idx = torch.cat([idx, torch.multinomial(probs, 1)], dim=1)
        return idx
