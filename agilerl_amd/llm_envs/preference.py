"""Preference (DPO) gyms.

Reference parity: ``agilerl/llm_envs/preference.py:21`` (PreferenceGym).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from .base import LLMEnvBase

__all__ = ["PreferenceGym", "SyntheticPreferenceGym"]


class SyntheticPreferenceGym(LLMEnvBase):
    """Token-space preference pairs: 'chosen' repeats the prompt's first
    token, 'rejected' is random noise — a separable preference signal."""

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

    def _batch(self) -> Dict[str, torch.Tensor]:
        B, P, C = self.data_batch_size, self.prompt_len, self.completion_len
        prompts = torch.from_numpy(self.rng.integers(1, self.vocab_size, (B, P), dtype=np.int64))
        chosen = torch.cat([prompts, prompts[:, :1].expand(B, C)], dim=1)
        rejected = torch.cat(
            [prompts, torch.from_numpy(self.rng.integers(1, self.vocab_size, (B, C), dtype=np.int64))],
            dim=1,
        )
        T = chosen.shape[1]
        pos = torch.arange(T - 1).unsqueeze(0)
        mask = (pos + 1 >= P).float().expand(B, T - 1)
        return {
            "chosen_ids": chosen,
            "rejected_ids": rejected,
            "chosen_mask": mask,
            "rejected_mask": mask.clone(),
        }

    def sample(self) -> Dict[str, torch.Tensor]:
        return self._batch()

    def sample_eval(self) -> Dict[str, torch.Tensor]:
        return self._batch()


class PreferenceGym(LLMEnvBase):
    """Text preference pairs (prompt, chosen, rejected) with a tokenizer."""

    def __init__(
        self,
        triples: List[Tuple[str, str, str]],
        tokenizer,
        data_batch_size: int = 8,
        max_tokens: int = 1024,
        seed: Optional[int] = None,
    ):
        self.triples = list(triples)
        self.tokenizer = tokenizer
        self.data_batch_size = data_batch_size
        self.max_tokens = max_tokens
        self.rng = np.random.default_rng(seed)
        if tokenizer.pad_token_id is None:
            tokenizer.pad_token = tokenizer.eos_token

    def _encode_side(self, prompts: List[str], completions: List[str]):
        tok = self.tokenizer
        rows, masks = [], []
        for p, c in zip(prompts, completions):
            p_ids = tok(p, add_special_tokens=False)["input_ids"]
            c_ids = tok(c, add_special_tokens=False)["input_ids"]
            ids = (p_ids + c_ids)[: self.max_tokens]
            m = ([0.0] * len(p_ids) + [1.0] * len(c_ids))[: self.max_tokens]
            rows.append(ids)
            masks.append(m)
        T = max(len(r) for r in rows)
        pad = tok.pad_token_id or 0
        ids = torch.full((len(rows), T), pad, dtype=torch.long)
        tgt = torch.zeros((len(rows), T), dtype=torch.float32)
        for i, (r, m) in enumerate(zip(rows, masks)):
            ids[i, : len(r)] = torch.tensor(r)
            tgt[i, : len(m)] = torch.tensor(m)
        return ids, tgt[:, 1:]

    def sample(self) -> Dict[str, torch.Tensor]:
        idx = self.rng.integers(0, len(self.triples), self.data_batch_size)
        sel = [self.triples[i] for i in idx]
        c_ids, c_mask = self._encode_side([t[0] for t in sel], [t[1] for t in sel])
        r_ids, r_mask = self._encode_side([t[0] for t in sel], [t[2] for t in sel])
        return {
            "chosen_ids": c_ids,
            "rejected_ids": r_ids,
            "chosen_mask": c_mask,
            "rejected_mask": r_mask,
        }

    def sample_eval(self) -> Dict[str, torch.Tensor]:
        return self.sample()
