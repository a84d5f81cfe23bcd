"""SFT — supervised fine-tuning with the fused CE path.

Reference parity: ``agilerl/algorithms/sft.py:37`` (supervised CE via the
fused linear kernel, packing support).  Loss = -mean masked token logprob,
computed through ``ops.fused_linear_logprobs`` so (B, T, V) logits are
never materialized.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from ... import ops
from ..core.registry import HyperparameterConfig
from .base import LLMAlgorithm

__all__ = ["SFT"]


class SFT(LLMAlgorithm):
    def __init__(
        self,
        model=None,
        model_config=None,
        model_name_or_path=None,
        tokenizer=None,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        lora_config=None,
        lr: float = 1e-5,
        micro_batch_size: int = 4,
        max_grad_norm: float = 1.0,
        dtype: torch.dtype = torch.bfloat16,
        gradient_checkpointing: bool = False,
        use_packing: bool = False,
        update_epochs: int = 1,
        seed=None,
        chunk_rows=None,
        activation_offload: bool = False,
        pad_token_id=None,
        device: str = "cpu",
        **kwargs,
    ):
        ref = self._resolve_reference_llm_kwargs(kwargs)
        model_name_or_path = ref.get("model_name_or_path", model_name_or_path)
        micro_batch_size = ref.get("micro_batch_size", micro_batch_size)
        use_packing = ref.get("use_packing", use_packing)
        update_epochs = ref.get("update_epochs", update_epochs)
        seed = ref.get("seed", seed)
        chunk_rows = ref.get("chunk_rows", chunk_rows)
        activation_offload = ref.get("activation_offload", activation_offload)
        pad_token_id = ref.get("pad_token_id", pad_token_id)
        if seed is not None:
            torch.manual_seed(int(seed))
        super().__init__(
            model=model, model_config=model_config, model_name_or_path=model_name_or_path,
            tokenizer=tokenizer, index=index, hp_config=hp_config, lora_config=lora_config,
            lr=lr, micro_batch_size=micro_batch_size, max_grad_norm=max_grad_norm,
            dtype=dtype, gradient_checkpointing=gradient_checkpointing, device=device,
            name="SFT",
        )
        self._accept_compat_kwargs(**kwargs)
        # padding-free grad pass (compute_logprobs_packed); opt-in as in GRPO
        self.use_packing = bool(use_packing)
        self.update_epochs = int(update_epochs)
        self.seed = seed
        self.chunk_rows = chunk_rows
        self.activation_offload = bool(activation_offload)
        if pad_token_id is not None and self.tokenizer is not None:
            self.tokenizer.pad_token_id = pad_token_id

    def learn(self, experiences: Dict[str, Any]) -> Dict[str, float]:
        """experiences: ids (B, T), attention_mask (B, T), action_mask
        (B, T-1) marking supervised target positions."""
        ids = experiences["ids"].to(self.device)
        attention_mask = experiences.get("attention_mask")
        attention_mask = (
            torch.ones_like(ids) if attention_mask is None else attention_mask.to(self.device)
        )
        action_mask = experiences["action_mask"].to(self.device).float()

        B = ids.shape[0]
        mb = max(self.micro_batch_size, 1)
        total_loss, n = 0.0, 0
        from ...llm.offload import activation_offload

        for _ in range(self.update_epochs):
            for s in range(0, B, mb):
                e = min(s + mb, B)
                fn = self.compute_logprobs_packed if self.use_packing else self.compute_logprobs
                with activation_offload(self.activation_offload):
                    logp = fn(ids[s:e], attention_mask[s:e], with_grad=True,
                              chunk_rows=self.chunk_rows)
                loss = -ops.masked_mean(logp, action_mask[s:e])
                self.backward_and_step(loss)
                total_loss += float(loss.detach())
                n += 1
        return {"loss": total_loss / max(n, 1)}

    def test(self, env, loop: int = 1, **kwargs) -> float:
        """Fitness: negative validation CE on one env batch."""
        losses = []
        for _ in range(loop):
            batch = env.sample_eval()
            with torch.no_grad():
                logp = self.compute_logprobs(
                    batch["ids"].to(self.device), batch["attention_mask"].to(self.device)
                )
                losses.append(float(-ops.masked_mean(logp, batch["action_mask"].to(self.device))))
        fitness = -float(np.mean(losses))
        self.fitness.append(fitness)
        return fitness
