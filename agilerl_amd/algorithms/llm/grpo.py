"""GRPO — group-relative policy optimization for LLM fine-tuning.

Reference parity: ``agilerl/algorithms/grpo.py:171`` — group-relative
advantage (:1219, here the HIP ``ops.group_advantage`` kernel), clipped
surrogate with optional k3 KL to the frozen base policy, token/sequence
loss normalization (:944/:1663), CISPO objective (:2051) via the
``cispo`` flag; GSPO (sequence-level IS) in ``gspo.py``.

Data flow per learn() call (one prompt batch of B*G completions):
  1. old/ref logprobs: fused no-grad pass (chunked lm_head kernels)
  2. advantages: group_advantage kernel over (B, G) rewards
  3. epochs x micro-batches: fused grad logprob pass -> fused
     token-masked surrogate kernel -> adapter-grad RCCL all-reduce ->
     AdamW on the LoRA params.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from ... import ops
from ...ops.grpo_loss import grpo_policy_loss
from ..core.registry import HyperparameterConfig
from .base import LLMAlgorithm

__all__ = ["GRPO"]


class GRPO(LLMAlgorithm):
    CISPO = False
    SEQUENCE_LEVEL_IS = False

    def __init__(
        self,
        model=None,
        model_config=None,
        model_name_or_path=None,
        tokenizer=None,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        lora_config=None,
        lr: float = 5e-6,
        group_size: int = 8,
        micro_batch_size: int = 2,
        update_epochs: int = 1,
        clip_coef: float = 0.2,
        clip_coef_lower: Optional[float] = None,
        beta: float = 0.0,
        scale_rewards: bool = True,
        loss_norm: str = "token",
        max_grad_norm: float = 1.0,
        temperature: float = 1.0,
        max_completion_tokens: int = 256,
        dtype: torch.dtype = torch.bfloat16,
        gradient_checkpointing: bool = False,
        use_packing: bool = False,
        generation: str = "hf",
        device: str = "cpu",
    ):
        super().__init__(
            model=model, model_config=model_config, model_name_or_path=model_name_or_path,
            tokenizer=tokenizer, index=index, hp_config=hp_config, lora_config=lora_config,
            lr=lr, micro_batch_size=micro_batch_size, max_grad_norm=max_grad_norm,
            temperature=temperature, max_completion_tokens=max_completion_tokens,
            dtype=dtype, gradient_checkpointing=gradient_checkpointing, device=device,
            name=type(self).__name__,
        )
        self.group_size = int(group_size)
        self.update_epochs = int(update_epochs)
        # padding-free grad/old-policy passes (compute_logprobs_packed);
        # opt-in until validated at 8-GPU scale
        self.use_packing = bool(use_packing)
        # "hf" = model.generate (default); "paged" = continuous-batching
        # paged-KV engine (llm/decode_engine.py, greedy-parity tested)
        if generation not in ("hf", "paged"):
            raise ValueError(f"generation must be 'hf' or 'paged', got {generation!r}")
        self.generation = generation
        self.clip_coef = float(clip_coef)
        self.clip_coef_lower = float(clip_coef_lower) if clip_coef_lower is not None else None
        self.beta = float(beta)  # k3 KL coefficient
        self.scale_rewards = bool(scale_rewards)
        self.loss_norm = loss_norm

    # ------------------------------------------------------------------
    def get_action(self, prompts: Dict[str, torch.Tensor], training: bool = True) -> torch.Tensor:
        """prompts: {"input_ids": (B, P), "attention_mask": (B, P)} already
        repeated group_size times by the env.  Returns full sequences
        (B, P+C)."""
        input_ids = prompts["input_ids"].to(self.device)
        attention_mask = prompts["attention_mask"].to(self.device)
        if self.generation == "paged":
            return self.generate_paged(input_ids, attention_mask, do_sample=training)
        return self.generate(input_ids, attention_mask, do_sample=training)

    # ------------------------------------------------------------------
    def _calculate_advantages(self, rewards: torch.Tensor) -> torch.Tensor:
        return ops.group_advantage(
            rewards.to(self.device).float(), self.group_size, scale=self.scale_rewards
        )

    def learn(self, experiences: Dict[str, Any]) -> Dict[str, float]:
        """experiences: ids (B, T) full prompt+completion token ids;
        action_mask (B, T-1) marking completion-token *targets*;
        rewards (B,) with B = n_prompts * group_size."""
        ids = experiences["ids"].to(self.device)
        attention_mask = experiences.get("attention_mask")
        if attention_mask is None:
            attention_mask = torch.ones_like(ids)
        attention_mask = attention_mask.to(self.device)
        action_mask = experiences["action_mask"].to(self.device).float()
        rewards = experiences["rewards"].to(self.device).float()

        self.check_seq_len_agreement(ids.shape[1])
        advantages = self._calculate_advantages(rewards)  # (B,)
        adv_tok = advantages.unsqueeze(1).expand_as(action_mask)

        B = ids.shape[0]
        mb = max(self.micro_batch_size, 1)
        logprob_fn = self.compute_logprobs_packed if self.use_packing else self.compute_logprobs

        # old-policy + reference logprobs (no grad, micro-batched)
        old_logp = torch.empty(action_mask.shape, device=self.device)
        ref_logp = torch.empty_like(old_logp) if self.beta > 0 else None
        for s in range(0, B, mb):
            e = min(s + mb, B)
            old_logp[s:e] = logprob_fn(ids[s:e], attention_mask[s:e], adapter="self")
            if ref_logp is not None:
                ref_logp[s:e] = logprob_fn(ids[s:e], attention_mask[s:e], adapter=None)

        clip_hi = 1.0 + self.clip_coef
        clip_lo = 1.0 - (self.clip_coef_lower if self.clip_coef_lower is not None else self.clip_coef)

        stats = {"loss": 0.0, "kl": 0.0, "clip_frac": 0.0}
        n_updates = 0
        for _ in range(self.update_epochs):
            perm = torch.randperm(B, device=self.device)
            for s in range(0, B, mb):
                sel = perm[s : s + mb]
                logp = logprob_fn(ids[sel], attention_mask[sel], with_grad=True)
                loss = self._policy_loss(
                    logp,
                    old_logp[sel],
                    adv_tok[sel],
                    action_mask[sel],
                    ref_logp[sel] if ref_logp is not None else None,
                    clip_lo,
                    clip_hi,
                )
                self.raise_if_loss_not_finite_on_any_rank(loss)
                self.backward_and_step(loss)
                with torch.no_grad():
                    ratio = (logp - old_logp[sel]).exp()
                    m = action_mask[sel].bool()
                    stats["loss"] += float(loss)
                    stats["clip_frac"] += float(
                        (((ratio > clip_hi) | (ratio < clip_lo)) & m).float().sum()
                        / m.float().sum().clamp(min=1)
                    )
                    if ref_logp is not None:
                        d = ref_logp[sel] - logp
                        stats["kl"] += float(ops.masked_mean(d.exp() - d - 1, m))
                n_updates += 1
        if n_updates:
            stats = {k: v / n_updates for k, v in stats.items()}
        stats["mean_reward"] = float(rewards.mean())
        return stats

    # ------------------------------------------------------------------
    def _policy_loss(self, logp, old_logp, adv_tok, mask, ref_logp, clip_lo, clip_hi):
        """Token-level surrogate (GRPO / CISPO) via the fused HIP kernel;
        GSPO overrides with the sequence-level pooled-IS objective."""
        return grpo_policy_loss(
            logp, old_logp, adv_tok, mask, ref_logp=ref_logp,
            clip_lo=clip_lo, clip_hi=clip_hi, kl_coef=self.beta,
            cispo=self.CISPO, loss_norm=self.loss_norm,
        )

    # ------------------------------------------------------------------
    def test(self, env, loop: int = 1, **kwargs) -> float:
        """Greedy-decode fitness: mean reward over one env batch."""
        rewards = []
        for _ in range(loop):
            prompts = env.reset()
            seqs = self.get_action(prompts, training=False)
            r = env.score(seqs)
            rewards.append(float(np.mean(r)))
        fitness = float(np.mean(rewards))
        self.fitness.append(fitness)
        return fitness
