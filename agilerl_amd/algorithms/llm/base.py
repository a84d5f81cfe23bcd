"""LLM fine-tuning algorithm base.

Reference parity: ``agilerl/algorithms/core/base.py:2415`` (LLMAlgorithm)
— HF model + LoRA adapters on an immutable base, fused logprob path,
adapter-only clone/evolution, adapter-directory checkpoints.  Redesigned
for one 8xMI355X node:

- ONE base model per process (frozen, bf16, resident in HBM — 288 GB
  makes QLoRA/ZeRO unnecessary for 8B); every
  population agent owns a named LoRA adapter slot inside the shared
  model, so clone = copy a few MB of adapter tensors and tournament
  weight transfer over xGMI is adapter-only (reference
  ``_resolve_clone_work_dir``/adapter temp-dir dance, base.py:3513).
- Logprobs never materialize (B, T, V): the decoder runs to hidden
  states and ``ops.fused_linear_logprobs`` (rocBLAS GEMM chunk + CDNA4
  logsumexp/gather kernels) produces token logprobs with exact backward.
- Data-parallel mode (one agent spanning N GPUs) synchronizes adapter
  gradients with a flat RCCL all-reduce (``parallel.allreduce_gradients``).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Optional

import numpy as np
import torch
import torch.nn as nn

from ...llm.lora import (
    LoraConfig,
    add_adapter,
    adapter_state_dict,
    apply_lora,
    iter_lora_modules,
    load_adapter,
    load_adapter_state_dict,
    mark_only_adapter_trainable,
    save_adapter,
    set_active_adapter,
)
from ...ops.fused_logprobs import fused_linear_logprobs
from ..core.base import EvolvableAlgorithm
from ..core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["LLMAlgorithm", "build_causal_lm"]


def build_causal_lm(
    model: Optional[nn.Module] = None,
    model_config: Optional[Any] = None,
    model_name_or_path: Optional[str] = None,
    dtype: torch.dtype = torch.bfloat16,
    device: str = "cpu",
) -> nn.Module:
    """Resolve a HF causal LM from an instance / config / local path.

    ``model_config`` may be a transformers PretrainedConfig or a plain dict
    (randomly initialized — the offline bench path; there is no network)."""
    if model is not None:
        return model
    from transformers import AutoConfig, AutoModelForCausalLM

    if model_config is not None:
        if isinstance(model_config, dict):
            cfg = dict(model_config)
            model_type = cfg.pop("model_type", "llama")
            model_config = AutoConfig.for_model(model_type, **cfg)
        lm = AutoModelForCausalLM.from_config(model_config, dtype=dtype)
    elif model_name_or_path is not None:
        lm = AutoModelForCausalLM.from_pretrained(model_name_or_path, dtype=dtype)
    else:
        raise ValueError("one of model / model_config / model_name_or_path is required")
    return lm.to(device)


class _AdapterView(nn.Module):
    """Exposes one agent's adapter parameters (mutation / optimizer surface)."""

    def __init__(self, model: nn.Module, adapter: str):
        super().__init__()
        object.__setattr__(self, "_ref_model", model)
        object.__setattr__(self, "_adapter", adapter)

    def parameters(self, recurse: bool = True):
        for _, module in iter_lora_modules(self._ref_model):
            if self._adapter in module.lora_A:
                yield module.lora_A[self._adapter]
                yield module.lora_B[self._adapter]

    def named_parameters(self, prefix: str = "", recurse: bool = True):
        for name, module in iter_lora_modules(self._ref_model):
            if self._adapter in module.lora_A:
                yield f"{name}.lora_A.{self._adapter}", module.lora_A[self._adapter]
                yield f"{name}.lora_B.{self._adapter}", module.lora_B[self._adapter]

    @property
    def mutation_methods(self) -> List[str]:
        return []


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr=RLParameter(min=1e-6, max=1e-3),
        beta=RLParameter(min=0.0, max=0.1),
    )


class LLMAlgorithm(EvolvableAlgorithm):
    def __init__(
        self,
        model: Optional[nn.Module] = None,
        model_config: Optional[Any] = None,
        model_name_or_path: Optional[str] = None,
        tokenizer: Optional[Any] = None,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        lora_config: Optional[Any] = None,
        lr: float = 5e-6,
        micro_batch_size: int = 2,
        max_grad_norm: float = 1.0,
        temperature: float = 1.0,
        max_completion_tokens: int = 256,
        dtype: torch.dtype = torch.bfloat16,
        gradient_checkpointing: bool = False,
        device: str = "cpu",
        name: Optional[str] = None,
    ):
        super().__init__(index=index, learn_step=1, device=device,
                         hp_config=hp_config or default_hp_config(), name=name or "LLM")
        self.lr = float(lr)
        self.micro_batch_size = int(micro_batch_size)
        self.max_grad_norm = float(max_grad_norm)
        self.temperature = float(temperature)
        self.max_completion_tokens = int(max_completion_tokens)
        self.dtype = dtype
        self.tokenizer = tokenizer
        self.gradient_checkpointing = gradient_checkpointing

        self.model = build_causal_lm(model, model_config, model_name_or_path, dtype, device)
        self.model.requires_grad_(False)
        if str(device).startswith("cuda") and dtype == torch.bfloat16:
            # fused CDNA4 RMSNorm + SwiGLU (analog of the reference's Liger
            # patches); GEMMs stay on hipBLASLt
            from ...architectures.llama_patches import (
                apply_hip_kernels_to_llama,
                patch_llama_swiglu,
            )

            apply_hip_kernels_to_llama(self.model)
            patch_llama_swiglu(self.model)
        if gradient_checkpointing and hasattr(self.model, "gradient_checkpointing_enable"):
            self.model.gradient_checkpointing_enable()

        if isinstance(lora_config, dict):
            lora_config = LoraConfig.from_dict(lora_config)
        self.lora_config = lora_config or LoraConfig()
        if not any(True for _ in iter_lora_modules(self.model)):
            apply_lora(self.model, self.lora_config, adapters=())
        self.adapter_name = f"agent_{index}"
        add_adapter(self.model, self.adapter_name)
        self._activate()

        self.optimizer = self._build_optimizer()
        self.register_network_group(NetworkGroup(eval_network="model", policy=True))
        self.register_optimizer(OptimizerConfig(name="optimizer", networks=["model"], lr_name="lr"))

    # ------------------------------------------------------------------
    # Registry overrides: the base model is shared and non-evolvable
    # ------------------------------------------------------------------
    def _registry_init(self) -> None:
        pass  # LLM agents do not follow the NetworkGroup module protocol

    @property
    def policy_network(self):
        return _AdapterView(self.model, self.adapter_name)

    @property
    def mutation_methods(self) -> List[str]:
        return []  # RL-HP mutations only (reference parity for LLM agents)

    def _build_optimizer(self):
        from ..core.optimizer_wrapper import OptimizerWrapper

        wrapper = OptimizerWrapper.__new__(OptimizerWrapper)
        wrapper.optimizer_cls = torch.optim.AdamW
        wrapper.lr = self.lr
        wrapper.network_names = ["model"]
        wrapper.lr_name = "lr"
        wrapper.optimizer_kwargs = {"weight_decay": 0.0}
        wrapper.multiagent = False
        params = list(self.policy_network.parameters())
        wrapper.optimizer = torch.optim.AdamW(params, lr=self.lr, weight_decay=0.0)
        return wrapper

    def _reinit_optimizers(self) -> None:
        self.optimizer = self._build_optimizer()

    def _activate(self, adapter: Optional[str] = "self") -> None:
        set_active_adapter(self.model, self.adapter_name if adapter == "self" else adapter)
        if adapter == "self":
            mark_only_adapter_trainable(self.model, self.adapter_name)

    # ------------------------------------------------------------------
    # Forward paths
    # ------------------------------------------------------------------
    def _decoder(self) -> nn.Module:
        m = self.model
        if hasattr(m, "get_decoder") and m.get_decoder() is not None:
            return m.get_decoder()
        if hasattr(m, "model"):
            return m.model
        raise RuntimeError("cannot locate decoder submodule")

    def _lm_head_weight(self) -> torch.Tensor:
        head = self.model.get_output_embeddings()
        return head.weight

    def compute_logprobs(
        self,
        input_ids: torch.Tensor,
        attention_mask: torch.Tensor,
        adapter: Optional[str] = "self",
        with_grad: bool = False,
        chunk_rows: Optional[int] = None,
    ) -> torch.Tensor:
        """Per-token logprobs of input_ids[t+1] given prefix: shape (B, T-1).

        ``adapter=None`` evaluates the frozen base model (the reference
        policy when no separate reference adapter exists)."""
        self._activate(adapter)
        ctx = torch.enable_grad() if with_grad else torch.no_grad()
        with ctx:
            hidden = self._decoder()(
                input_ids=input_ids, attention_mask=attention_mask
            ).last_hidden_state
            targets = input_ids[:, 1:]
            lp = fused_linear_logprobs(
                hidden[:, :-1, :], self._lm_head_weight(), targets,
                temperature=self.temperature, chunk_rows=chunk_rows,
            )
        if adapter != "self":
            self._activate("self")
        return lp

    def compute_logprobs_packed(
        self,
        input_ids: torch.Tensor,
        attention_mask: torch.Tensor,
        adapter: Optional[str] = "self",
        with_grad: bool = False,
        chunk_rows: Optional[int] = None,
    ) -> torch.Tensor:
        """Padding-free variant of compute_logprobs (SURVEY §2.8 long-context
        strategy; reference llm_packing.py): real tokens are concatenated
        into one row with per-sequence position ids and a block-diagonal
        causal mask, so the decoder and the chunked lm_head GEMMs process
        only real tokens.  Returns (B, T-1) with zeros at pad positions —
        numerically equivalent to compute_logprobs on the real positions
        (RoPE attention depends only on position differences, so the
        per-sequence position restart is offset-invariant)."""
        from ...llm.packing import pack_padded_batch

        pack = pack_padded_batch(input_ids, attention_mask)
        packed, pos, cu = pack["packed_ids"], pack["position_ids"], pack["cu_seqlens"]
        N = packed.shape[1]
        lengths = cu[1:] - cu[:-1]
        seq_id = torch.repeat_interleave(
            torch.arange(lengths.numel(), device=packed.device), lengths
        )
        same = seq_id.unsqueeze(0) == seq_id.unsqueeze(1)
        causal = torch.arange(N, device=packed.device).unsqueeze(0) <= torch.arange(
            N, device=packed.device
        ).unsqueeze(1)
        dtype = next(self.model.parameters()).dtype
        neg = torch.finfo(dtype).min
        mask4d = torch.where(same & causal, 0.0, neg).to(dtype)[None, None]

        self._activate(adapter)
        ctx = torch.enable_grad() if with_grad else torch.no_grad()
        with ctx:
            hidden = self._decoder()(
                input_ids=packed, position_ids=pos, attention_mask=mask4d
            ).last_hidden_state[0]  # (N, H)
            # positions whose next token is real and in the same sequence
            k = torch.arange(N - 1, device=packed.device)
            valid = seq_id[k] == seq_id[k + 1]
            vk = k[valid]
            lp_valid = fused_linear_logprobs(
                hidden[vk], self._lm_head_weight(), packed[0, vk + 1],
                temperature=self.temperature, chunk_rows=chunk_rows,
            )
            B, T = input_ids.shape
            out = lp_valid.new_zeros(B * T)
            out[pack["indices"][vk]] = lp_valid
            out = out.reshape(B, T)[:, :-1]
        if adapter != "self":
            self._activate("self")
        return out

    def _ensure_engine(self, tokens_needed: int, B: int):
        from ...llm.decode_engine import DecodeEngine

        engine = getattr(self, "_decode_engine", None)
        page_size = 16
        if engine is None or engine.cache.num_pages * page_size < tokens_needed:
            engine = DecodeEngine(
                self.model,
                max_batch=max(B, 8),
                num_pages=max(64, 2 * (tokens_needed // page_size + 1)),
                page_size=page_size,
                eos_token_id=getattr(self.model.config, "eos_token_id", None),
                set_adapter_fn=self._activate,
            )
            self._decode_engine = engine
        engine.max_batch = max(engine.max_batch, B)
        return engine

    @torch.no_grad()
    def stream_generate(
        self,
        input_ids: torch.Tensor,
        attention_mask: torch.Tensor,
        max_new_tokens: Optional[int] = None,
        do_sample: bool = True,
        temperature: Optional[float] = None,
    ):
        """Yield completion token ids one step at a time (single prompt)
        through the paged engine — the serving /generate/stream source."""
        C = max_new_tokens or self.max_completion_tokens
        ids = input_ids.reshape(-1)
        engine = self._ensure_engine(int(ids.numel()) + C, 1)
        self.model.eval()
        temp = temperature if temperature is not None else self.temperature
        sid = engine.submit(
            ids, max_new_tokens=C,
            temperature=temp if do_sample else 0.0, adapter="self",
        )
        try:
            seen = 0
            done = False
            while engine.has_work() and not done:
                finished = engine.step()
                done = any(fsid == sid for fsid, _ in finished)
                seq = engine.active.get(sid)
                gen = (
                    seq.generated if seq is not None
                    else next(ids_ for fsid, ids_ in finished if fsid == sid)[
                        input_ids.reshape(-1).numel():
                    ].tolist()
                )
                for tok in list(gen)[seen:]:
                    yield int(tok)
                seen = len(gen)
        finally:
            engine.cancel(sid)
            self.model.train()

    @torch.no_grad()
    def generate_paged(
        self,
        input_ids: torch.Tensor,
        attention_mask: torch.Tensor,
        max_new_tokens: Optional[int] = None,
        do_sample: bool = True,
        temperature: Optional[float] = None,
    ) -> torch.Tensor:
        """Generation through the continuous-batching paged-KV engine
        (llm/decode_engine.py).  Same (B, P+C) rectangular contract as
        ``generate``; completions are right-padded with pad_token_id."""
        C = max_new_tokens or self.max_completion_tokens
        B, P = input_ids.shape
        pad_id = getattr(self.model.config, "pad_token_id", 0) or 0
        tokens_needed = int(attention_mask.sum()) + B * C
        engine = self._ensure_engine(tokens_needed, B)
        self.model.eval()
        temp = temperature if temperature is not None else self.temperature
        # Merged generation: fold this agent's LoRA delta into the base
        # weights for the whole phase, so decode runs plain GEMMs (no
        # adapter side-ops, ~3 fewer kernels per projection per step) and
        # the decode graph is keyed/captured adapter-free.  Unmerge
        # restores the base bits EXACTLY from a snapshot (llm/lora.py).
        merged = bool(getattr(self, "merged_generation", True))
        if merged:
            from ...llm.lora import merge_adapter, set_active_adapter, unmerge_adapter

            merge_adapter(self.model, self.adapter_name)
            set_active_adapter(self.model, None)
        try:
            sids = []
            for i in range(B):
                row_mask = attention_mask[i].bool()
                sids.append(engine.submit(
                    input_ids[i][row_mask], max_new_tokens=C,
                    temperature=temp if do_sample else 0.0,
                    adapter=None if merged else "self",
                ))
            results = engine.run_all()
        finally:
            if merged:
                unmerge_adapter(self.model)
                self._activate("self")
        self.model.train()
        out = torch.full((B, P + C), pad_id, dtype=torch.long, device=self.device)
        out[:, :P] = input_ids
        # behavior-policy logprobs on the TARGET grid (B, P+C-1): position j
        # holds the sampling logprob of token out[:, j+1]; zero elsewhere
        # (the reference keeps vLLM sampling logprobs the same way)
        samp = torch.zeros((B, P + C - 1), dtype=torch.float32, device=self.device)
        for i, sid in enumerate(sids):
            comp = results[sid][int(attention_mask[i].sum()):]
            out[i, P : P + comp.numel()] = comp.to(self.device)
            lps = engine.finished_logps.get(sid)
            if lps:
                n = min(len(lps), C)
                samp[i, P - 1 : P - 1 + n] = torch.tensor(
                    lps[:n], dtype=torch.float32, device=self.device
                )
        self.last_sampling_logps = samp
        return out

    @torch.no_grad()
    def generate(
        self,
        input_ids: torch.Tensor,
        attention_mask: torch.Tensor,
        max_new_tokens: Optional[int] = None,
        do_sample: bool = True,
        temperature: Optional[float] = None,
    ) -> torch.Tensor:
        """Batched KV-cached generation with the agent's adapter active."""
        self._activate("self")
        self.last_sampling_logps = None  # only the paged engine captures these
        self.model.eval()
        pad_id = getattr(self.model.config, "pad_token_id", None)
        if pad_id is None:
            pad_id = getattr(self.model.config, "eos_token_id", 0) or 0
            if isinstance(pad_id, (list, tuple)):
                pad_id = pad_id[0]
        gen_kwargs = dict(getattr(self, "sampling_kwargs", {}) or {})
        gen_kwargs.setdefault("top_p", 1.0)
        gen_kwargs.setdefault("top_k", 0)
        out = self.model.generate(
            input_ids=input_ids,
            attention_mask=attention_mask,
            max_new_tokens=max_new_tokens or self.max_completion_tokens,
            do_sample=do_sample,
            temperature=temperature or self.temperature,
            pad_token_id=pad_id,
            use_cache=True,
            **gen_kwargs,
        )
        self.model.train()
        return out

    # ------------------------------------------------------------------
    # Cross-rank safety (reference base.py:5201-5222 / grpo.py:727-740)
    # ------------------------------------------------------------------
    def raise_if_loss_not_finite_on_any_rank(self, loss: torch.Tensor) -> None:
        """All DP ranks must raise together, or the healthy ranks hang in
        the next collective while one rank has already aborted."""
        from ...parallel import DistributedState

        state = DistributedState.get()
        bad = (~torch.isfinite(loss.detach())).float().reshape(1)
        if state.is_distributed:
            import torch.distributed as dist

            flag = bad.to(self.device) if state.backend == "nccl" else bad.cpu()
            dist.all_reduce(flag, op=dist.ReduceOp.MAX)
            bad = flag
        if float(bad) > 0:
            raise RuntimeError(
                f"non-finite loss detected on at least one rank (local loss={float(loss)})"
            )

    def check_seq_len_agreement(self, seq_len: int) -> None:
        """DP micro-batches must agree on T or the flat all-reduce deadlocks
        with mismatched bucket sizes; fail fast with a clear error."""
        from ...parallel import DistributedState

        state = DistributedState.get()
        if not state.is_distributed:
            return
        import torch.distributed as dist

        t = torch.tensor([float(seq_len), -float(seq_len)])
        if state.backend == "nccl":
            t = t.to(self.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        mx, mn = float(t[0]), -float(t[1])
        if mx != mn:
            raise RuntimeError(
                f"sequence-length mismatch across ranks: min={int(mn)} max={int(mx)} "
                f"(local={seq_len}); pad batches to a common length"
            )

    # ------------------------------------------------------------------
    # Reference-API kwargs shared by the LLM algorithms
    # ------------------------------------------------------------------
    _REF_IGNORED_KWARGS = (
        "quantization_config", "use_memory_efficient_params",
        "reduce_memory_peak", "calc_position_embeddings",
        "cast_logprobs_to_fp32", "hf_generate_chunk_size",
        "lora_target_scope", "use_separate_reference_adapter", "clone",
        "actor_network", "max_model_len", "batch_size", "mini_batch_size",
        "pad_token",
    )

    @classmethod
    def _resolve_reference_llm_kwargs(cls, kwargs: Dict[str, Any]) -> Dict[str, Any]:
        """Pop reference-spelling kwargs from ``kwargs`` and return the
        recognized values under their native names.  Infra-only keys
        (quantization, vLLM memory gymnastics — unnecessary on 288 GB
        HBM3E) warn and are dropped; anything left in ``kwargs`` falls
        through to ``_accept_compat_kwargs`` (TypeError on true typos)."""
        import warnings

        out: Dict[str, Any] = {}
        for ref, ours in (("model_name", "model_name_or_path"),
                          ("micro_batch_size_per_gpu", "micro_batch_size"),
                          ("use_sequence_packing", "use_packing"),
                          ("max_output_tokens", "max_completion_tokens")):
            if ref in kwargs:
                out[ours] = kwargs.pop(ref)
        for k in ("seed", "chunk_rows", "pad_token_id", "activation_offload",
                  "use_liger_loss", "update_epochs", "nll_alpha"):
            if k in kwargs:
                out[k] = kwargs.pop(k)
        ignored = [k for k in cls._REF_IGNORED_KWARGS if kwargs.pop(k, None) is not None]
        if ignored:
            warnings.warn(
                f"{cls.__name__} ignores reference-only kwargs {ignored} "
                "(see docs/llm_finetuning.md for the MI355X equivalents)",
                RuntimeWarning,
            )
        return out

    def backward_and_step(self, loss: torch.Tensor, accumulate: bool = False) -> None:
        loss.backward()
        if accumulate:
            return
        from ...parallel import DistributedState

        state = DistributedState.get()
        if state.is_distributed:
            from ...parallel.ddp import allreduce_gradients

            allreduce_gradients(self.policy_network)
        params = list(self.policy_network.parameters())
        torch.nn.utils.clip_grad_norm_(params, self.max_grad_norm)
        self.optimizer.step()
        self.optimizer.zero_grad(set_to_none=True)

    # ------------------------------------------------------------------
    # Evolution: adapter-only clone / checkpoint
    # ------------------------------------------------------------------
    def clone(self, index: Optional[int] = None, wrap: bool = True) -> "LLMAlgorithm":
        import copy as _copy

        clone = _copy.copy(self)  # shallow: shares the base model
        # The decode engine is bound to the PARENT's _activate (its
        # set_adapter_fn) — sharing it would make the clone's paged
        # generation sample under the parent's LoRA adapter.  Drop it so the
        # clone lazily rebuilds an engine bound to its own _activate.
        clone.__dict__.pop("_decode_engine", None)
        clone.registry = self.registry
        new_index = self.index if index is None else index
        clone.index = new_index
        clone.adapter_name = f"agent_{new_index}_{id(clone) & 0xFFFF:x}"
        add_adapter(self.model, clone.adapter_name, init=False)
        load_adapter_state_dict(
            self.model, clone.adapter_name, adapter_state_dict(self.model, self.adapter_name)
        )
        for attr in ("fitness", "scores", "steps"):
            setattr(clone, attr, list(getattr(self, attr)))
        clone.optimizer = clone._build_optimizer()
        return clone

    def clean_up(self) -> None:
        """Remove this agent's adapter from the shared model."""
        for _, module in iter_lora_modules(self.model):
            if self.adapter_name in module.lora_A:
                module.lora_A.pop(self.adapter_name)
                module.lora_B.pop(self.adapter_name)
            if module.active_adapter == self.adapter_name:
                module.active_adapter = None

    # ------------------------------------------------------------------
    def get_checkpoint_dict(self) -> Dict[str, Any]:
        return {
            "agilerl_version": "0.1.0",
            "algo": self.algo,
            "adapter_state": adapter_state_dict(self.model, self.adapter_name),
            "lora_config": self.lora_config.to_dict(),
            "attributes": self.inspect_attributes(ignore=("tokenizer", "adapter_name")),
        }

    def _apply_checkpoint(self, ckpt: Dict[str, Any]) -> None:
        load_adapter_state_dict(self.model, self.adapter_name, ckpt["adapter_state"])
        for k, v in ckpt["attributes"].items():
            if k in ("device", "dtype", "adapter_name"):
                continue
            setattr(self, k, v)
        self.optimizer = self._build_optimizer()

    def save_checkpoint(self, path: str) -> None:
        """Directory checkpoint: <path>/actor/{adapter_model.safetensors,
        adapter_config.json} + attributes.pt (reference layout)."""
        os.makedirs(path, exist_ok=True)
        # temporarily expose this agent's adapter under the canonical name
        state = adapter_state_dict(self.model, self.adapter_name)
        from safetensors.torch import save_file

        actor_dir = os.path.join(path, "actor")
        os.makedirs(actor_dir, exist_ok=True)
        save_file(state, os.path.join(actor_dir, "adapter_model.safetensors"))
        import json

        with open(os.path.join(actor_dir, "adapter_config.json"), "w") as f:
            json.dump(self.lora_config.to_dict(), f, indent=2)
        torch.save(
            self.inspect_attributes(ignore=("tokenizer", "adapter_name")),
            os.path.join(path, "attributes.pt"),
        )

    def load_checkpoint(self, path: str) -> None:
        load_adapter(self.model, self.adapter_name, os.path.join(path, "actor"))
        attrs = torch.load(os.path.join(path, "attributes.pt"), weights_only=False)
        for k, v in attrs.items():
            if k in ("device", "dtype"):
                continue
            setattr(self, k, v)
        self.optimizer = self._build_optimizer()

    # ------------------------------------------------------------------
    @classmethod
    def population(cls, size: int, *args, **kwargs) -> List["LLMAlgorithm"]:
        """Agent 0 builds the base model; the rest are adapter clones."""
        first = cls(*args, index=0, **kwargs)
        agents = [first]
        for i in range(1, size):
            agents.append(first.clone(index=i))
        return agents
