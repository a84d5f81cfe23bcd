"""LLM environment base + experience assembly.

Reference parity: ``agilerl/llm_envs/base.py:93`` (HuggingFaceGym — HF
dataset + rank-sharded DataLoader, chat templating) — reshaped for this
framework: an LLM env yields *token batches* ready for the model
(prompts repeated ``group_size`` times), scores generated sequences, and
the loop assembles the (ids, action_mask, rewards) experiences dict the
GRPO-family ``learn()`` consumes.
"""

from __future__ import annotations

from typing import Dict

import numpy as np
import torch

__all__ = ["LLMEnvBase", "HuggingFaceGym", "make_grpo_experiences"]


class LLMEnvBase:
    """Interface: reset() -> prompt batch; score(sequences) -> rewards."""

    group_size: int = 1
    prompt_len: int = 0  # padded prompt length of the last reset batch

    def reset(self) -> Dict[str, torch.Tensor]:
        raise NotImplementedError

    def score(self, sequences: torch.Tensor) -> np.ndarray:
        raise NotImplementedError

    def step(self, sequences: torch.Tensor):
        """Single-turn default: scoring ends the episode."""
        rewards = self.score(sequences)
        return rewards, True


def make_grpo_experiences(
    env: LLMEnvBase,
    sequences: torch.Tensor,
    rewards: np.ndarray,
    pad_token_id: int = 0,
) -> Dict[str, torch.Tensor]:
    """Assemble the GRPO learn() payload from generated sequences.

    ``action_mask[b, j]`` marks target position j (predicting token j+1)
    as a completion token: j+1 >= prompt_len and ids[b, j+1] != pad.
    """
    ids = sequences
    B, T = ids.shape
    P = env.prompt_len
    attention_mask = (ids != pad_token_id).long()
    # left-padded prompts: everything from position P on is completion
    pos = torch.arange(T - 1, device=ids.device).unsqueeze(0)
    action_mask = (pos + 1 >= P) & (ids[:, 1:] != pad_token_id)
    return {
        "ids": ids,
        "attention_mask": attention_mask,
        "action_mask": action_mask.float(),
        "rewards": torch.as_tensor(np.asarray(rewards), dtype=torch.float32),
    }


class HuggingFaceGym(LLMEnvBase):
    """HF-datasets-backed prompt environment with epoch dataloaders.

    Reference parity: ``agilerl/llm_envs/base.py:93`` — binds a
    ``datasets.Dataset`` train/test pair behind shuffled epoch
    DataLoaders, shards batches across DP ranks (the reference uses
    accelerate-prepared loaders; here a first-party index shard over
    ``DistributedState``), renders prompts through the tokenizer's chat
    template, and exposes the evaluation-mode toggle + epoch counter.

    Single-turn contract (like :class:`ReasoningGym`): ``reset()`` yields
    the next tokenized prompt batch repeated ``group_size`` times;
    ``score(sequences)`` applies ``reward_fn(completion_text, answer)``.
    """

    def __init__(
        self,
        train_dataset,
        test_dataset,
        tokenizer,
        reward_fn,
        prompt_key: str = "question",
        answer_key: str = "answer",
        data_batch_size: int = 8,
        group_size: int = 4,
        max_prompt_tokens: int = 512,
        apply_chat_template: bool = True,
        system_prompt=None,
        shard_across_ranks: bool = True,
        seed: int = 42,
        conversation_template=None,
        data_batch_size_per_gpu: int = None,
        max_context_length: int = None,
        min_completion_length: int = None,
        accelerator=None,
    ):
        # reference llm_envs/base.py:103-106 spellings: per-GPU batch is the
        # same thing here (one process per GPU); conversation_template is a
        # list of {role, content} messages prefixed before each prompt;
        # max_context_length caps the tokenized prompt
        if data_batch_size_per_gpu is not None:
            data_batch_size = data_batch_size_per_gpu
        if max_context_length is not None:
            max_prompt_tokens = int(max_context_length)
        self.conversation_template = conversation_template
        self.min_completion_length = min_completion_length
        if accelerator is not None:
            import warnings

            warnings.warn(
                "HuggingFaceGym ignores `accelerator`: rank sharding reads "
                "torch.distributed directly.", RuntimeWarning,
            )
        self.tokenizer = tokenizer
        self.reward_fn = reward_fn
        self.prompt_key = prompt_key
        self.answer_key = answer_key
        self.data_batch_size = int(data_batch_size)
        self.group_size = int(group_size)
        self.max_prompt_tokens = int(max_prompt_tokens)
        self.apply_chat_template = apply_chat_template
        self.system_prompt = system_prompt
        self.seed = seed
        if tokenizer.pad_token_id is None:
            tokenizer.pad_token = tokenizer.eos_token

        rank, world = 0, 1
        if shard_across_ranks:
            from ..parallel import DistributedState

            state = DistributedState.get()
            rank, world = state.rank, state.world_size
        self._rank, self._world = rank, world
        self._datasets = {"train": train_dataset, "test": test_dataset}
        self.dataset_size = {k: len(v) for k, v in self._datasets.items()}
        self.num_epochs = 0
        self.evaluation_mode = False
        self._rng = np.random.default_rng(seed)
        self._iters = {"train": self._epoch_iter("train"), "test": self._epoch_iter("test")}
        self._batch_answers = []
        self.prompt_len = 0

    # ------------------------------------------------------------------
    def _shard_indices(self, split: str) -> np.ndarray:
        n = self.dataset_size[split]
        idx = np.arange(n)
        if split == "train":
            self._rng.shuffle(idx)
        if self._world > 1 and n >= self._world:
            idx = idx[self._rank :: self._world]  # disjoint per-rank shard
        return idx

    def _epoch_iter(self, split: str):
        while True:
            idx = self._shard_indices(split)
            for s in range(0, len(idx), self.data_batch_size):
                chunk = idx[s : s + self.data_batch_size]
                if len(chunk) == 0:
                    continue
                yield [self._datasets[split][int(i)] for i in chunk]
            if split == "train":
                self.num_epochs += 1

    def eval(self, on: bool = True) -> None:
        """Toggle evaluation mode (test split, unshuffled)."""
        self.evaluation_mode = bool(on)

    def reset(self, reset_dataloaders: bool = False) -> Dict[str, torch.Tensor]:
        if reset_dataloaders:
            self._rng = np.random.default_rng(self.seed)
            self._iters = {
                "train": self._epoch_iter("train"),
                "test": self._epoch_iter("test"),
            }
            self.num_epochs = 0
        split = "test" if self.evaluation_mode else "train"
        rows = next(self._iters[split])
        texts = [self._render(str(r[self.prompt_key])) for r in rows]
        self._batch_answers = [
            r.get(self.answer_key) for r in rows for _ in range(self.group_size)
        ]
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

    def _render(self, prompt: str) -> str:
        if self.apply_chat_template and getattr(self.tokenizer, "chat_template", None):
            messages = []
            if self.conversation_template:
                # reference base.py:37: few-shot / instruction turns prefixed
                # before each prompt
                messages.extend(dict(m) for m in self.conversation_template)
            if self.system_prompt and not any(
                m.get("role") == "system" for m in messages
            ):
                messages.insert(0, {"role": "system", "content": self.system_prompt})
            messages.append({"role": "user", "content": prompt})
            return self.tokenizer.apply_chat_template(
                messages, tokenize=False, add_generation_prompt=True
            )
        if self.conversation_template:
            prefix = "\n".join(
                f"{m.get('role', 'user').capitalize()}: {m.get('content', '')}"
                for m in self.conversation_template
            )
            return f"{prefix}\nUser: {prompt}"
        return prompt

    def score(self, sequences: torch.Tensor) -> np.ndarray:
        completions = self.tokenizer.batch_decode(
            sequences[:, self.prompt_len :], skip_special_tokens=True
        )
        rewards = [
            float(self.reward_fn(c, a))
            for c, a in zip(completions, self._batch_answers)
        ]
        return np.asarray(rewards, dtype=np.float32)
