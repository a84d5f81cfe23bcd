"""DPO — direct preference optimization.

Reference parity: ``agilerl/algorithms/dpo.py:38`` (preference pairs,
fused path via ``llm_ops/fused_loss.py:740`` LigerDPOWithAlpha; here the
per-token logprobs come from the fused CDNA4 kernels and the pairwise
sigmoid loss is elementwise).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch
import torch.nn.functional as F

from ... import ops
from ..core.registry import HyperparameterConfig
from .base import LLMAlgorithm

__all__ = ["DPO"]


class DPO(LLMAlgorithm):
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
        beta: float = 0.1,
        label_smoothing: float = 0.0,
        micro_batch_size: int = 2,
        max_grad_norm: float = 1.0,
        dtype: torch.dtype = torch.bfloat16,
        gradient_checkpointing: bool = False,
        use_packing: bool = False,
        nll_alpha: float = 0.0,
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
        nll_alpha = ref.get("nll_alpha", nll_alpha)
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
            name="DPO",
        )
        self._accept_compat_kwargs(**kwargs)
        self.beta = float(beta)
        self.label_smoothing = float(label_smoothing)
        # reference fused_loss.py:740 LigerDPOWithAlpha: nll_alpha adds an
        # SFT-style NLL term on the chosen completions to the DPO loss
        self.nll_alpha = float(nll_alpha)
        self.seed = seed
        self.chunk_rows = chunk_rows
        self.activation_offload = bool(activation_offload)
        # padding-free logprob passes (compute_logprobs_packed); opt-in
        self.use_packing = bool(use_packing)
        if pad_token_id is not None and self.tokenizer is not None:
            self.tokenizer.pad_token_id = pad_token_id

    def _seq_logp(self, ids, attention_mask, action_mask, with_grad: bool, adapter="self"):
        from ...llm.offload import activation_offload

        fn = self.compute_logprobs_packed if self.use_packing else self.compute_logprobs
        with activation_offload(self.activation_offload and with_grad):
            logp = fn(ids, attention_mask, adapter=adapter, with_grad=with_grad,
                      chunk_rows=self.chunk_rows)
        return (logp * action_mask).sum(dim=1)

    def learn(self, experiences: Dict[str, Any]) -> Dict[str, float]:
        """experiences: chosen_ids/rejected_ids (B, T), *_attention_mask,
        chosen_mask/rejected_mask (B, T-1)."""
        dev = self.device
        c_ids = experiences["chosen_ids"].to(dev)
        r_ids = experiences["rejected_ids"].to(dev)
        c_am = experiences.get("chosen_attention_mask", torch.ones_like(c_ids)).to(dev)
        r_am = experiences.get("rejected_attention_mask", torch.ones_like(r_ids)).to(dev)
        c_mask = experiences["chosen_mask"].to(dev).float()
        r_mask = experiences["rejected_mask"].to(dev).float()

        B = c_ids.shape[0]
        mb = max(self.micro_batch_size, 1)
        stats = {"loss": 0.0, "margin": 0.0, "accuracy": 0.0}
        n = 0
        for s in range(0, B, mb):
            e = min(s + mb, B)
            with torch.no_grad():
                ref_c = self._seq_logp(c_ids[s:e], c_am[s:e], c_mask[s:e], False, adapter=None)
                ref_r = self._seq_logp(r_ids[s:e], r_am[s:e], r_mask[s:e], False, adapter=None)
            pol_c = self._seq_logp(c_ids[s:e], c_am[s:e], c_mask[s:e], True)
            pol_r = self._seq_logp(r_ids[s:e], r_am[s:e], r_mask[s:e], True)
            logits = self.beta * ((pol_c - ref_c) - (pol_r - ref_r))
            loss = (
                -F.logsigmoid(logits) * (1 - self.label_smoothing)
                - F.logsigmoid(-logits) * self.label_smoothing
            ).mean()
            if self.nll_alpha > 0:
                # reference LigerDPOWithAlpha (fused_loss.py:740): SFT-style
                # NLL on the chosen completions stabilizes DPO
                nll = -(pol_c / c_mask[s:e].sum(dim=1).clamp(min=1.0)).mean()
                loss = loss + self.nll_alpha * nll
            self.backward_and_step(loss)
            stats["loss"] += float(loss.detach())
            stats["margin"] += float(logits.detach().mean())
            stats["accuracy"] += float((logits.detach() > 0).float().mean())
            n += 1
        return {k: v / max(n, 1) for k, v in stats.items()}

    def test(self, env, loop: int = 1, **kwargs) -> float:
        """Fitness: preference accuracy on an eval batch."""
        accs = []
        for _ in range(loop):
            b = env.sample_eval()
            with torch.no_grad():
                dev = self.device
                pol_c = self._seq_logp(
                    b["chosen_ids"].to(dev), torch.ones_like(b["chosen_ids"]).to(dev),
                    b["chosen_mask"].to(dev).float(), False,
                )
                pol_r = self._seq_logp(
                    b["rejected_ids"].to(dev), torch.ones_like(b["rejected_ids"]).to(dev),
                    b["rejected_mask"].to(dev).float(), False,
                )
                ref_c = self._seq_logp(
                    b["chosen_ids"].to(dev), torch.ones_like(b["chosen_ids"]).to(dev),
                    b["chosen_mask"].to(dev).float(), False, adapter=None,
                )
                ref_r = self._seq_logp(
                    b["rejected_ids"].to(dev), torch.ones_like(b["rejected_ids"]).to(dev),
                    b["rejected_mask"].to(dev).float(), False, adapter=None,
                )
                logits = (pol_c - ref_c) - (pol_r - ref_r)
                accs.append(float((logits > 0).float().mean()))
        fitness = float(np.mean(accs))
        self.fitness.append(fitness)
        return fitness
