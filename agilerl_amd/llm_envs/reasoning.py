"""Single-turn reasoning gyms (prompt -> completion -> reward).

Reference parity: ``agilerl/llm_envs/reasoning.py:23`` (ReasoningGym:
dataset batch -> completions -> reward_fn).  Two variants:

- :class:`ReasoningGym` — text prompts + tokenizer + text reward_fn
  (offline datasets / lists; there is no hub access in this environment).
- :class:`TokenReasoningGym` — operates purely in token space with a
  token-level reward_fn; used for synthetic benchmarks (BASELINE config 5
  runs on random-init Llama-3-8B with synthetic prompt/response tokens)
  and for the tiny-model test fixtures.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional

import numpy as np
import torch

from .base import LLMEnvBase

__all__ = ["ReasoningGym", "TokenReasoningGym"]


class TokenReasoningGym(LLMEnvBase):
    """Synthetic token-space reasoning env.

    Prompts are random token windows; the default reward is the fraction
    of completion tokens equal to the prompt's FIRST token — a learnable
    copy task with dense signal, used by tests and the offline GRPO bench.
    """

    def __init__(
        self,
        vocab_size: int,
        prompt_len: int = 32,
        data_batch_size: int = 4,
        group_size: int = 4,
        reward_fn: Optional[Callable[[torch.Tensor, int], np.ndarray]] = None,
        pad_token_id: int = 0,
        seed: Optional[int] = None,
    ):
        self.vocab_size = vocab_size
        self.prompt_len = int(prompt_len)
        self.data_batch_size = int(data_batch_size)
        self.group_size = int(group_size)
        self.reward_fn = reward_fn
        self.pad_token_id = pad_token_id
        self.rng = np.random.default_rng(seed)
        self._last_prompts: Optional[torch.Tensor] = None

    def reset(self) -> Dict[str, torch.Tensor]:
        B, P = self.data_batch_size, self.prompt_len
        prompts = torch.from_numpy(
            self.rng.integers(1, self.vocab_size, size=(B, P), dtype=np.int64)
        )
        prompts = prompts.repeat_interleave(self.group_size, dim=0)
        self._last_prompts = prompts
        return {
            "input_ids": prompts,
            "attention_mask": torch.ones_like(prompts),
        }

    def score(self, sequences: torch.Tensor) -> np.ndarray:
        completions = sequences[:, self.prompt_len :]
        if self.reward_fn is not None:
            return np.asarray(self.reward_fn(sequences, self.prompt_len), dtype=np.float32)
        target = self._last_prompts[:, 0].to(sequences.device)
        if completions.numel() == 0:
            return np.zeros(sequences.shape[0], dtype=np.float32)
        match = (completions == target.unsqueeze(1)).float().mean(dim=1)
        return match.cpu().numpy().astype(np.float32)


class ReasoningGym(LLMEnvBase):
    """Text reasoning env over an offline prompt/answer list or dataset."""

    def __init__(
        self,
        prompts: List[str],
        answers: Optional[List[Any]],
        reward_fn: Callable[[str, Any], float],
        tokenizer,
        data_batch_size: int = 4,
        group_size: int = 4,
        max_prompt_tokens: int = 512,
        apply_chat_template: bool = True,
        system_prompt: Optional[str] = None,
        seed: Optional[int] = None,
        shard_across_ranks: bool = True,
    ):
        prompts = list(prompts)
        answers = list(answers) if answers is not None else [None] * len(prompts)
        if shard_across_ranks:
            # per-rank data sharding (reference: accelerate-prepared
            # DataLoader split, llm_envs/base.py:156) — each DP rank sees a
            # disjoint prompt slice so the population never re-scores the
            # same prompt twice in one pass
            from ..parallel import DistributedState

            state = DistributedState.get()
            if state.world_size > 1 and len(prompts) >= state.world_size:
                prompts = prompts[state.rank :: state.world_size]
                answers = answers[state.rank :: state.world_size]
        self.prompts = prompts
        self.answers = answers
        self.reward_fn = reward_fn
        self.tokenizer = tokenizer
        self.data_batch_size = int(data_batch_size)
        self.group_size = int(group_size)
        self.max_prompt_tokens = int(max_prompt_tokens)
        self.apply_chat_template = apply_chat_template
        self.system_prompt = system_prompt
        self.rng = np.random.default_rng(seed)
        self._batch_answers: List[Any] = []
        self.prompt_len = 0
        if tokenizer.pad_token_id is None:
            tokenizer.pad_token = tokenizer.eos_token

    def _render(self, prompt: str) -> str:
        if self.apply_chat_template and getattr(self.tokenizer, "chat_template", None):
            messages = []
            if self.system_prompt:
                messages.append({"role": "system", "content": self.system_prompt})
            messages.append({"role": "user", "content": prompt})
            return self.tokenizer.apply_chat_template(
                messages, tokenize=False, add_generation_prompt=True
            )
        return prompt

    def reset(self) -> Dict[str, torch.Tensor]:
        idx = self.rng.integers(0, len(self.prompts), size=self.data_batch_size)
        texts = [self._render(self.prompts[i]) for i in idx]
        self._batch_answers = [self.answers[i] for i in idx for _ in range(self.group_size)]
        enc = self.tokenizer(
            texts,
            return_tensors="pt",
            padding=True,
            truncation=True,
            max_length=self.max_prompt_tokens,
            padding_side="left",
        )
        input_ids = enc["input_ids"].repeat_interleave(self.group_size, dim=0)
        attention_mask = enc["attention_mask"].repeat_interleave(self.group_size, dim=0)
        self.prompt_len = input_ids.shape[1]
        return {"input_ids": input_ids, "attention_mask": attention_mask}

    def score(self, sequences: torch.Tensor) -> np.ndarray:
        completions = self.tokenizer.batch_decode(
            sequences[:, self.prompt_len :], skip_special_tokens=True
        )
        rewards = [
            float(self.reward_fn(c, a)) for c, a in zip(completions, self._batch_answers)
        ]
        return np.asarray(rewards, dtype=np.float32)

    @classmethod
    def from_dataset(
        cls,
        dataset,
        reward_fn,
        tokenizer,
        prompt_key: str = "question",
        answer_key: str = "answer",
        **kwargs,
    ) -> "ReasoningGym":
        """Build from a HF ``datasets.Dataset``, a local saved-dataset path,
        or a list of dicts (reference HuggingFaceGym, llm_envs/base.py:93;
        offline only — no hub access)."""
        if isinstance(dataset, str):
            from datasets import load_from_disk

            dataset = load_from_disk(dataset)
        if hasattr(dataset, "column_names"):
            prompts = list(dataset[prompt_key])
            answers = list(dataset[answer_key]) if answer_key in dataset.column_names else None
        else:
            prompts = [row[prompt_key] for row in dataset]
            answers = [row.get(answer_key) for row in dataset]
        return cls(prompts, answers, reward_fn, tokenizer, **kwargs)
