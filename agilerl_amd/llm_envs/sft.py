"""SFT gyms (supervised batches).

Reference parity: ``agilerl/llm_envs/sft.py:21`` (SFTGym).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from .base import LLMEnvBase

__all__ = ["SFTGym", "SyntheticSFTGym"]


class SyntheticSFTGym(LLMEnvBase):
    """Token-space SFT data: completion = prompt's first token repeated
    (a learnable supervised pattern for tests/benchmarks)."""

    def __init__(
        self,
        vocab_size: int,
        prompt_len: int = 16,
        completion_len: int = 16,
        data_batch_size: int = 8,
        seed: Optional[int] = None,
    ):
        self.vocab_size = vocab_size
        self.prompt_len = prompt_len
        self.completion_len = completion_len
        self.data_batch_size = data_batch_size
        self.rng = np.random.default_rng(seed)

    def _batch(self, batch_size: int) -> Dict[str, torch.Tensor]:
        B, P, C = batch_size, self.prompt_len, self.completion_len
        prompts = torch.from_numpy(self.rng.integers(1, self.vocab_size, (B, P), dtype=np.int64))
        completion = prompts[:, :1].expand(B, C)
        ids = torch.cat([prompts, completion], dim=1)
        T = ids.shape[1]
        pos = torch.arange(T - 1).unsqueeze(0)
        action_mask = (pos + 1 >= P).float().expand(B, T - 1)
        return {
            "ids": ids,
            "attention_mask": torch.ones_like(ids),
            "action_mask": action_mask,
        }

    def sample(self) -> Dict[str, torch.Tensor]:
        return self._batch(self.data_batch_size)

    def sample_eval(self) -> Dict[str, torch.Tensor]:
        return self._batch(self.data_batch_size)


class SFTGym(LLMEnvBase):
    """Text SFT over (prompt, completion) pairs with a tokenizer."""

    def __init__(
        self,
        pairs: List[Tuple[str, str]],
        tokenizer,
        data_batch_size: int = 8,
        max_tokens: int = 1024,
        eval_fraction: float = 0.1,
        seed: Optional[int] = None,
    ):
        self.pairs = list(pairs)
        self.tokenizer = tokenizer
        self.data_batch_size = data_batch_size
        self.max_tokens = max_tokens
        self.rng = np.random.default_rng(seed)
        n_eval = max(1, int(len(self.pairs) * eval_fraction))
        self.eval_pairs = self.pairs[:n_eval]
        self.train_pairs = self.pairs[n_eval:] or self.pairs
        if tokenizer.pad_token_id is None:
            tokenizer.pad_token = tokenizer.eos_token

    def _encode(self, pairs) -> Dict[str, torch.Tensor]:
        tok = self.tokenizer
        rows, masks = [], []
        for prompt, completion in pairs:
            p_ids = tok(prompt, add_special_tokens=False)["input_ids"]
            c_ids = tok(completion, add_special_tokens=False)["input_ids"]
            ids = (p_ids + c_ids)[: self.max_tokens]
            mask = ([0.0] * len(p_ids) + [1.0] * len(c_ids))[: self.max_tokens]
            rows.append(ids)
            masks.append(mask)
        T = max(len(r) for r in rows)
        pad = tok.pad_token_id or 0
        ids = torch.full((len(rows), T), pad, dtype=torch.long)
        am = torch.zeros((len(rows), T), dtype=torch.long)
        tgt = torch.zeros((len(rows), T), dtype=torch.float32)
        for i, (r, m) in enumerate(zip(rows, masks)):
            ids[i, : len(r)] = torch.tensor(r)
            am[i, : len(r)] = 1
            tgt[i, : len(m)] = torch.tensor(m)
        # action_mask over target positions (predicting ids[:, 1:])
        action_mask = tgt[:, 1:]
        return {"ids": ids, "attention_mask": am, "action_mask": action_mask}

    def sample(self) -> Dict[str, torch.Tensor]:
        idx = self.rng.integers(0, len(self.train_pairs), self.data_batch_size)
        return self._encode([self.train_pairs[i] for i in idx])

    def sample_eval(self) -> Dict[str, torch.Tensor]:
        idx = self.rng.integers(0, len(self.eval_pairs), min(self.data_batch_size, len(self.eval_pairs)))
        return self._encode([self.eval_pairs[i] for i in idx])
