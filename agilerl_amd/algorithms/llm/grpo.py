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
        importance_sampling_level: str = "token",
        advantage_level: str = "trajectory",
        sampling_is_correction: bool = False,
        sampling_is_cap: float = 2.0,
        grad_accumulation_steps: int = 1,
        max_grad_norm: float = 1.0,
        temperature: float = 1.0,
        max_completion_tokens: int = 256,
        dtype: torch.dtype = torch.bfloat16,
        gradient_checkpointing: bool = False,
        use_packing="auto",
        generation: str = "hf",
        loss_type: str = "grpo",
        adv_norm: Optional[str] = None,
        whiten_advantages: bool = False,
        adv_clip_range: Optional[float] = None,
        filter_zero_adv: bool = False,
        adv_filter_eps: float = 0.0,
        use_kl_advantage_shaping: bool = False,
        top_k: Optional[int] = None,
        top_p: Optional[float] = None,
        min_p: Optional[float] = None,
        repetition_penalty: Optional[float] = None,
        min_output_tokens: Optional[int] = None,
        seed: Optional[int] = None,
        chunk_rows: Optional[int] = None,
        cosine_lr_schedule_config=None,
        activation_offload: bool = False,
        pad_token_id: Optional[int] = None,
        use_liger_loss: bool = True,
        device: str = "cpu",
        **kwargs,
    ):
        # ------------------------------------------------------------------
        # Reference kwarg aliases (reference grpo.py:441-539) — accepted so
        # code written against the reference constructs GRPO unchanged
        # ------------------------------------------------------------------
        if "model_name" in kwargs:
            model_name_or_path = model_name_or_path or kwargs.pop("model_name")
        if "use_sequence_packing" in kwargs:
            use_packing = kwargs.pop("use_sequence_packing")
        if "vllm_importance_sampling_correction" in kwargs:
            sampling_is_correction = kwargs.pop("vllm_importance_sampling_correction")
        if "vllm_importance_sampling_cap" in kwargs:
            sampling_is_cap = kwargs.pop("vllm_importance_sampling_cap")
        if "micro_batch_size_per_gpu" in kwargs:
            micro_batch_size = kwargs.pop("micro_batch_size_per_gpu")
        if "action_granularity" in kwargs:
            importance_sampling_level = kwargs.pop("action_granularity")
        if "advantage_granularity" in kwargs:
            advantage_level = kwargs.pop("advantage_granularity")
        if "max_output_tokens" in kwargs:
            max_completion_tokens = kwargs.pop("max_output_tokens")
        if kwargs.pop("use_vllm", False) or kwargs.pop("vllm_config", None) is not None:
            import warnings

            warnings.warn(
                "use_vllm/vllm_config map to the native paged-KV decode "
                "engine on MI355X (generation='paged'); vLLM itself is not "
                "used.",
                RuntimeWarning,
            )
            generation = "paged"
        # infra kwargs with no MI355X role: warn-and-ignore (288 GB HBM
        # removes the memory gymnastics; no bitsandbytes/DeepSpeed/vLLM)
        _ignored = [k for k in (
            "quantization_config", "use_memory_efficient_params",
            "reduce_memory_peak", "calc_position_embeddings",
            "cast_logprobs_to_fp32", "hf_generate_chunk_size",
            "lora_target_scope", "use_separate_reference_adapter",
            "turn_advantage_trajectory_fallback", "clone", "actor_network",
            "max_model_len", "batch_size", "mini_batch_size", "pad_token",
        ) if kwargs.pop(k, None) is not None]
        if _ignored:
            import warnings

            warnings.warn(
                f"GRPO ignores reference-only kwargs {_ignored} (see "
                "docs/llm_finetuning.md for the MI355X equivalents)",
                RuntimeWarning,
            )
        if adv_norm is not None:
            # reference grpo.py:453 adv_norm: "mean_std" | "mean_only"
            if adv_norm not in ("mean_std", "mean_only"):
                raise ValueError("adv_norm must be 'mean_std' or 'mean_only'")
            scale_rewards = adv_norm == "mean_std"
        if seed is not None:
            torch.manual_seed(int(seed))
        super().__init__(
            model=model, model_config=model_config, model_name_or_path=model_name_or_path,
            tokenizer=tokenizer, index=index, hp_config=hp_config, lora_config=lora_config,
            lr=lr, micro_batch_size=micro_batch_size, max_grad_norm=max_grad_norm,
            temperature=temperature, max_completion_tokens=max_completion_tokens,
            dtype=dtype, gradient_checkpointing=gradient_checkpointing, device=device,
            name=type(self).__name__,
        )
        self._accept_compat_kwargs(**kwargs)
        if loss_type not in ("grpo", "gspo", "cispo"):
            raise ValueError("loss_type must be grpo | gspo | cispo")
        self.loss_type = loss_type
        if loss_type == "cispo":
            self.CISPO = True
        elif loss_type == "gspo":
            importance_sampling_level = "trajectory"
        self.adv_norm = adv_norm or ("mean_std" if scale_rewards else "mean_only")
        self.whiten_advantages = bool(whiten_advantages)
        self.adv_clip_range = adv_clip_range
        self.filter_zero_adv = bool(filter_zero_adv)
        self.adv_filter_eps = float(adv_filter_eps)
        self.use_kl_advantage_shaping = bool(use_kl_advantage_shaping)
        self.seed = seed
        self.chunk_rows = chunk_rows
        self.activation_offload = bool(activation_offload)
        self.use_liger_loss = bool(use_liger_loss)  # False forces the eager loss path
        # sampling controls forwarded to generation (reference top_k/top_p/
        # min_p/repetition_penalty/min_output_tokens)
        self.sampling_kwargs = {
            k: v for k, v in {
                "top_k": top_k, "top_p": top_p, "min_p": min_p,
                "repetition_penalty": repetition_penalty,
                "min_new_tokens": min_output_tokens,
            }.items() if v is not None
        }
        if pad_token_id is not None and self.tokenizer is not None:
            self.tokenizer.pad_token_id = pad_token_id
        self.lr_scheduler = None
        if cosine_lr_schedule_config is not None:
            from ...llm.scheduler import create_warmup_cosine_scheduler

            cfg = cosine_lr_schedule_config
            total = int(cfg.get("num_epochs", 1) if isinstance(cfg, dict)
                        else getattr(cfg, "num_epochs", 1))
            warm = float(cfg.get("warmup_proportion", 0.03) if isinstance(cfg, dict)
                         else getattr(cfg, "warmup_proportion", 0.03))
            self.lr_scheduler = create_warmup_cosine_scheduler(
                self.optimizer, total_steps=max(total, 1), warmup_ratio=warm
            )
        self.group_size = int(group_size)
        self.update_epochs = int(update_epochs)
        # padding-free grad/old-policy passes (compute_logprobs_packed):
        # True / False / "auto" (default) — auto packs whenever the batch
        # carries meaningful padding (>10% pad tokens), since packing only
        # pays when there are pads to skip (GPU-equivalence tested)
        if use_packing not in (True, False, "auto"):
            raise ValueError(f"use_packing must be bool or 'auto', got {use_packing!r}")
        self.use_packing = use_packing
        # "hf" = model.generate (default); "paged" = continuous-batching
        # paged-KV engine (llm/decode_engine.py, greedy-parity tested)
        if generation not in ("hf", "paged"):
            raise ValueError(f"generation must be 'hf' or 'paged', got {generation!r}")
        self.generation = generation
        if isinstance(clip_coef, (list, tuple)):
            # reference CISPO configs give RATIO bounds `clip_coef: [lo, hi]`
            # (cispo_quant_bench.yaml); our epsilons are hi-1 / 1-lo
            lo, hi = (float(x) for x in clip_coef)
            clip_coef = hi - 1.0
            if clip_coef_lower is None:
                clip_coef_lower = 1.0 - lo
        self.clip_coef = float(clip_coef)
        self.clip_coef_lower = float(clip_coef_lower) if clip_coef_lower is not None else None
        self.beta = float(beta)  # k3 KL coefficient
        self.scale_rewards = bool(scale_rewards)
        if loss_norm == "micro_batch":
            # reference models/algorithms/grpo.py:45 spelling: each
            # micro-batch normalized by its own action-token count — our
            # "token" mode
            loss_norm = "token"
        if loss_norm not in ("token", "sequence", "accumulation_window"):
            raise ValueError(f"unknown loss_norm {loss_norm!r}")
        self.loss_norm = loss_norm
        # importance-sampling pooling level (reference grpo.py:1848-1903):
        # "token" (GRPO) / "turn" (per-turn pooled ratio) / "trajectory"
        # (GSPO).  Subclass default via SEQUENCE_LEVEL_IS for back-compat.
        if importance_sampling_level not in ("token", "turn", "trajectory"):
            raise ValueError(
                f"unknown importance_sampling_level {importance_sampling_level!r}"
            )
        if self.SEQUENCE_LEVEL_IS and importance_sampling_level == "token":
            importance_sampling_level = "trajectory"
        self.importance_sampling_level = importance_sampling_level
        # advantage granularity: "trajectory" (group-relative over final
        # rewards) or "turn" (group-relative per turn over turn_rewards,
        # broadcast to that turn's tokens — reference grpo.py:1250-1379)
        if advantage_level not in ("trajectory", "turn"):
            raise ValueError(f"unknown advantage_level {advantage_level!r}")
        self.advantage_level = advantage_level
        # truncated-IS correction against the decode engine's sampling
        # logprobs (the vLLM-IS analog, reference grpo.py:2500-2510)
        self.sampling_is_correction = bool(sampling_is_correction)
        self.sampling_is_cap = float(sampling_is_cap)
        self.grad_accumulation_steps = max(1, int(grad_accumulation_steps))

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
        turn_ids = experiences.get("turn_ids")
        if turn_ids is not None:
            turn_ids = turn_ids.to(self.device)
        sampling_logps = experiences.get("sampling_logps")
        if sampling_logps is not None and self.sampling_is_correction:
            sampling_logps = sampling_logps.to(self.device).float()
        else:
            sampling_logps = None
        turn_rewards = experiences.get("turn_rewards")

        self.check_seq_len_agreement(ids.shape[1])
        if (
            self.advantage_level == "turn"
            and turn_rewards is not None
            and turn_ids is not None
        ):
            adv_tok = self._turn_advantages(
                turn_rewards.to(self.device).float(), turn_ids, action_mask
            )
        else:
            advantages = self._calculate_advantages(rewards)  # (B,)
            adv_tok = advantages.unsqueeze(1).expand_as(action_mask)
        if self.filter_zero_adv:
            # reference grpo.py:461: drop samples with ~zero advantage
            samp_adv = (adv_tok * action_mask).sum(1) / action_mask.sum(1).clamp(min=1)
            action_mask = action_mask * (samp_adv.abs() > self.adv_filter_eps).float().unsqueeze(1)
        if self.whiten_advantages:
            m = action_mask.bool()
            vals = adv_tok[m]
            if vals.numel() > 1:
                adv_tok = torch.where(
                    m, (adv_tok - vals.mean()) / (vals.std() + 1e-8),
                    torch.zeros_like(adv_tok),
                )
        if self.adv_clip_range is not None:
            adv_tok = adv_tok.clamp(-self.adv_clip_range, self.adv_clip_range)

        B = ids.shape[0]
        mb = max(self.micro_batch_size, 1)
        packing = self.use_packing
        if packing == "auto":
            pad_frac = 1.0 - attention_mask.float().mean().item()
            packing = pad_frac > 0.10
        _logprob_fn = self.compute_logprobs_packed if packing else self.compute_logprobs

        def logprob_fn(*a, **kw):
            kw.setdefault("chunk_rows", self.chunk_rows)
            return _logprob_fn(*a, **kw)

        # old-policy + reference logprobs (no grad, micro-batched)
        old_logp = torch.empty(action_mask.shape, device=self.device)
        ref_logp = torch.empty_like(old_logp) if self.beta > 0 else None
        for s in range(0, B, mb):
            e = min(s + mb, B)
            old_logp[s:e] = logprob_fn(ids[s:e], attention_mask[s:e], adapter="self")
            if ref_logp is not None:
                ref_logp[s:e] = logprob_fn(ids[s:e], attention_mask[s:e], adapter=None)
        if self.use_kl_advantage_shaping and ref_logp is not None:
            # ART-style zero-mean KL shaping (reference grpo.py:1488-1503):
            # k3 KL of the pre-update policy vs the reference adapter
            d = ref_logp - old_logp
            kl = (d.exp() - d - 1) * action_mask
            avg = kl.sum(-1, keepdim=True) / action_mask.sum(-1, keepdim=True).clamp(min=1.0)
            adv_tok = adv_tok + self.beta * (avg - kl)

        clip_hi = 1.0 + self.clip_coef
        clip_lo = 1.0 - (self.clip_coef_lower if self.clip_coef_lower is not None else self.clip_coef)

        stats = {"loss": 0.0, "kl": 0.0, "clip_frac": 0.0}
        n_updates = 0
        accum = self.grad_accumulation_steps
        for _ in range(self.update_epochs):
            perm = torch.randperm(B, device=self.device)
            starts = list(range(0, B, mb))
            # accumulation windows: optimizer steps every `accum`
            # micro-batches; "accumulation_window" loss_norm divides every
            # micro-batch by the WINDOW's action-token count (reference
            # grpo.py:1619-1694 loss-norm windows)
            for i, s in enumerate(starts):
                sel = perm[s : s + mb]
                window_denom = None
                if self.loss_norm == "accumulation_window":
                    w0 = (i // accum) * accum
                    wsel = perm[starts[w0] : starts[w0] + mb * accum]
                    window_denom = float(action_mask[wsel].sum().clamp(min=1.0))
                from ...llm.offload import activation_offload

                with activation_offload(self.activation_offload):
                    logp = logprob_fn(ids[sel], attention_mask[sel], with_grad=True)
                loss = self._policy_loss(
                    logp,
                    old_logp[sel],
                    adv_tok[sel],
                    action_mask[sel],
                    ref_logp[sel] if ref_logp is not None else None,
                    clip_lo,
                    clip_hi,
                    turn_ids=turn_ids[sel] if turn_ids is not None else None,
                    sampling_logp=sampling_logps[sel] if sampling_logps is not None else None,
                    denom_tokens=window_denom,
                )
                self.raise_if_loss_not_finite_on_any_rank(loss)
                accumulate = (i % accum) != accum - 1 and i != len(starts) - 1
                self.backward_and_step(loss, accumulate=accumulate)
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
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        if n_updates:
            stats = {k: v / n_updates for k, v in stats.items()}
        stats["mean_reward"] = float(rewards.mean())
        return stats

    # ------------------------------------------------------------------
    def _turn_advantages(
        self,
        turn_rewards: torch.Tensor,  # (B, K) per-turn rewards
        turn_ids: torch.Tensor,      # (B, T) turn index per target (-1 pad)
        action_mask: torch.Tensor,   # (B, T)
    ) -> torch.Tensor:
        """Group-relative advantage per (group, turn), broadcast to each
        turn's tokens (reference grpo.py:1250-1379)."""
        B, K = turn_rewards.shape
        g = turn_rewards.view(-1, self.group_size, K)
        adv = g - g.mean(dim=1, keepdim=True)
        if self.scale_rewards:
            adv = adv / (g.std(dim=1, keepdim=True) + 1e-8)
        adv = adv.view(B, K)
        safe = turn_ids.clamp(min=0, max=K - 1).long()
        return adv.gather(1, safe) * (turn_ids >= 0).float() * action_mask

    def _policy_loss(self, logp, old_logp, adv_tok, mask, ref_logp, clip_lo,
                     clip_hi, turn_ids=None, sampling_logp=None,
                     denom_tokens=None):
        """Surrogate at the configured IS level: token level (GRPO/CISPO)
        runs the fused HIP kernel; turn/trajectory pooling and the
        sampling-IS correction take the eager path (pooling is cheap; the
        logprob computation upstream stays fused either way)."""
        return grpo_policy_loss(
            logp, old_logp, adv_tok, mask, ref_logp=ref_logp,
            clip_lo=clip_lo, clip_hi=clip_hi, kl_coef=self.beta,
            cispo=self.CISPO, loss_norm=self.loss_norm,
            level=self.importance_sampling_level, turn_ids=turn_ids,
            sampling_logp=sampling_logp, sampling_cap=self.sampling_is_cap,
            denom_tokens=denom_tokens,
            force_eager=not getattr(self, "use_liger_loss", True),
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
