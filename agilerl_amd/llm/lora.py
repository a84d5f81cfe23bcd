"""First-party LoRA (no PEFT dependency).

The reference uses HF PEFT adapters named ``actor`` / ``reference`` /
``critic`` on an immutable base model (``base.py:4344``, ``_allowed_adapters``
:2546) and checkpoints each adapter as ``adapter_model.safetensors`` +
``adapter_config.json``.  This module reimplements that
surface directly on top of ``nn.Linear``:

- :class:`LoraLinear` holds the frozen base weight plus one (A, B) pair
  per adapter name; the active adapter is selected per-module (fast
  whole-model switch via :func:`set_active_adapter`).
- Merged-weight math: ``y = x W^T + scaling * (x A^T) B^T`` with
  ``scaling = lora_alpha / r``.
- :func:`apply_lora` walks a HF model and wraps the target projections;
  :func:`save_adapter` / :func:`load_adapter` use the reference's on-disk
  adapter-directory layout.
"""

from __future__ import annotations

import json
import math
import os
import re
from typing import Dict, Iterable, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = [
    "LoraConfig",
    "LoraLinear",
    "apply_lora",
    "set_active_adapter",
    "add_adapter",
    "adapter_state_dict",
    "load_adapter_state_dict",
    "save_adapter",
    "load_adapter",
    "mark_only_adapter_trainable",
    "iter_lora_modules",
]

DEFAULT_TARGETS = ["q_proj", "k_proj", "v_proj", "o_proj", "gate_proj", "up_proj", "down_proj"]


class LoraConfig:
    def __init__(
        self,
        r: int = 16,
        lora_alpha: int = 32,
        lora_dropout: float = 0.0,
        target_modules: Optional[List[str]] = None,
        bias: str = "none",
        task_type: str = "CAUSAL_LM",
    ):
        self.r = int(r)
        self.lora_alpha = int(lora_alpha)
        self.lora_dropout = float(lora_dropout)
        self.target_modules = list(target_modules or DEFAULT_TARGETS)
        self.bias = bias
        self.task_type = task_type

    def to_dict(self) -> Dict:
        return {
            "r": self.r,
            "lora_alpha": self.lora_alpha,
            "lora_dropout": self.lora_dropout,
            "target_modules": self.target_modules,
            "bias": self.bias,
            "task_type": self.task_type,
            "peft_type": "LORA",
        }

    @classmethod
    def from_dict(cls, d: Dict) -> "LoraConfig":
        return cls(
            r=d.get("r", 16),
            lora_alpha=d.get("lora_alpha", 32),
            lora_dropout=d.get("lora_dropout", 0.0),
            target_modules=d.get("target_modules"),
            bias=d.get("bias", "none"),
            task_type=d.get("task_type", "CAUSAL_LM"),
        )

    def __eq__(self, other) -> bool:
        return isinstance(other, LoraConfig) and self.to_dict() == other.to_dict()


class LoraLinear(nn.Module):
    """nn.Linear with per-adapter low-rank deltas."""

    def __init__(self, base: nn.Linear, config: LoraConfig, adapters: Iterable[str] = ("actor",)):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad = False
        self.config = config
        self.scaling = config.lora_alpha / config.r
        self.dropout = nn.Dropout(config.lora_dropout) if config.lora_dropout > 0 else None
        self.lora_A = nn.ParameterDict()
        self.lora_B = nn.ParameterDict()
        self.active_adapter = None
        for name in adapters:
            self.add_adapter(name)
        if self.active_adapter is None and self.lora_A:
            self.active_adapter = next(iter(self.lora_A))

    @property
    def in_features(self) -> int:
        return self.base.in_features

    @property
    def out_features(self) -> int:
        return self.base.out_features

    def add_adapter(self, name: str, init: bool = True) -> None:
        if name in self.lora_A:
            return
        dev = self.base.weight.device
        dt = self.base.weight.dtype
        A = torch.zeros(self.config.r, self.base.in_features, device=dev, dtype=dt)
        B = torch.zeros(self.base.out_features, self.config.r, device=dev, dtype=dt)
        if init:
            nn.init.kaiming_uniform_(A, a=math.sqrt(5))
        self.lora_A[name] = nn.Parameter(A)
        self.lora_B[name] = nn.Parameter(B)
        if self.active_adapter is None:
            self.active_adapter = name

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.base(x)
        name = self.active_adapter
        if name is not None and name in self.lora_A and self._merged is None:
            h = self.dropout(x) if self.dropout is not None else x
            out = out + F.linear(F.linear(h, self.lora_A[name]), self.lora_B[name]) * self.scaling
        return out

    # ------------------------------------------------------------------
    # Merge for generation: fold B@A*scale into the base weight so decode
    # runs plain GEMMs (no adapter side-ops).  The pre-merge bf16 weight
    # bits are snapshotted, so unmerge restores the base EXACTLY (a
    # subtract would drift by rounding each cycle).  Base weights are
    # frozen under LoRA training, so the snapshot is always valid.
    # ------------------------------------------------------------------
    _merged = None  # (adapter_name, snapshot) while merged

    @torch.no_grad()
    def merge_adapter(self, name: str) -> None:
        if self._merged is not None:
            raise RuntimeError(f"already merged: {self._merged[0]}")
        if name not in self.lora_A:
            return
        w = self.base.weight
        snapshot = w.detach().clone()
        delta = (self.lora_B[name].float() @ self.lora_A[name].float()) * self.scaling
        w.data.copy_((w.float() + delta).to(w.dtype))
        self._merged = (name, snapshot)

    @torch.no_grad()
    def unmerge_adapter(self) -> None:
        if self._merged is None:
            return
        _, snapshot = self._merged
        self.base.weight.data.copy_(snapshot)
        self._merged = None


# ---------------------------------------------------------------------------
# Model-level helpers
# ---------------------------------------------------------------------------

def iter_lora_modules(model: nn.Module):
    for name, module in model.named_modules():
        if isinstance(module, LoraLinear):
            yield name, module


def apply_lora(
    model: nn.Module,
    config: LoraConfig,
    adapters: Iterable[str] = ("actor",),
) -> nn.Module:
    """Wrap every target nn.Linear in the model with LoraLinear (in place)."""
    targets = set(config.target_modules)

    def _wrap(target_names) -> int:
        wrapped = 0
        for parent_name, parent in list(model.named_modules()):
            for child_name, child in list(parent.named_children()):
                if isinstance(child, nn.Linear) and child_name in target_names:
                    setattr(parent, child_name, LoraLinear(child, config, adapters))
                    wrapped += 1
        return wrapped

    if _wrap(targets) == 0:
        # reference configs for wrapped/multimodal models spell targets as
        # `q_proj.linear`; retry with the leading segment against plain
        # decoder layouts before giving up
        stripped = {t.split(".", 1)[0] for t in targets}
        if stripped != targets and _wrap(stripped) > 0:
            import warnings

            warnings.warn(
                f"LoRA target_modules {sorted(targets)} matched nothing; "
                f"matched the base names {sorted(stripped)} instead",
                RuntimeWarning,
            )
        else:
            raise ValueError(
                f"LoRA target_modules {sorted(targets)} matched no nn.Linear "
                "in the model — adapters would be empty"
            )
    mark_only_adapter_trainable(model)
    return model


def mark_only_adapter_trainable(model: nn.Module, adapter: Optional[str] = None) -> None:
    for name, p in model.named_parameters():
        if ".lora_A." in name or ".lora_B." in name:
            p.requires_grad = adapter is None or name.split(".")[-1] == adapter
        else:
            p.requires_grad = False


def set_active_adapter(model: nn.Module, adapter: Optional[str]) -> None:
    """adapter=None disables all deltas (pure base model = reference policy
    when no separate reference adapter is used)."""
    for _, module in iter_lora_modules(model):
        module.active_adapter = adapter if (adapter is None or adapter in module.lora_A) else None


def add_adapter(model: nn.Module, name: str, init: bool = True) -> None:
    for _, module in iter_lora_modules(model):
        module.add_adapter(name, init=init)


def adapter_state_dict(model: nn.Module, adapter: str) -> Dict[str, torch.Tensor]:
    out = {}
    for name, module in iter_lora_modules(model):
        if adapter in module.lora_A:
            out[f"{name}.lora_A.weight"] = module.lora_A[adapter].detach().cpu()
            out[f"{name}.lora_B.weight"] = module.lora_B[adapter].detach().cpu()
    return out


@torch.no_grad()
def load_adapter_state_dict(model: nn.Module, adapter: str, state: Dict[str, torch.Tensor]) -> None:
    for name, module in iter_lora_modules(model):
        ka, kb = f"{name}.lora_A.weight", f"{name}.lora_B.weight"
        if ka in state:
            module.add_adapter(adapter, init=False)
            module.lora_A[adapter].copy_(state[ka].to(module.lora_A[adapter].device))
            module.lora_B[adapter].copy_(state[kb].to(module.lora_B[adapter].device))


def save_adapter(model: nn.Module, adapter: str, directory: str, config: Optional[LoraConfig] = None) -> None:
    """Reference-compatible adapter dir: adapter_model.safetensors + adapter_config.json."""
    os.makedirs(directory, exist_ok=True)
    state = adapter_state_dict(model, adapter)
    from safetensors.torch import save_file

    save_file(state, os.path.join(directory, "adapter_model.safetensors"))
    cfg = config
    if cfg is None:
        for _, module in iter_lora_modules(model):
            cfg = module.config
            break
    with open(os.path.join(directory, "adapter_config.json"), "w") as f:
        json.dump(cfg.to_dict() if cfg else {}, f, indent=2)


def load_adapter(model: nn.Module, adapter: str, directory: str, strict_config: bool = True) -> None:
    from safetensors.torch import load_file

    cfg_path = os.path.join(directory, "adapter_config.json")
    if strict_config and os.path.exists(cfg_path):
        with open(cfg_path) as f:
            saved = LoraConfig.from_dict(json.load(f))
        for _, module in iter_lora_modules(model):
            live = module.config
            if (live.r, live.lora_alpha) != (saved.r, saved.lora_alpha):
                raise ValueError(
                    f"LoRA config mismatch: checkpoint (r={saved.r}, alpha={saved.lora_alpha}) "
                    f"vs model (r={live.r}, alpha={live.lora_alpha})"
                )
            break
    state = load_file(os.path.join(directory, "adapter_model.safetensors"))
    load_adapter_state_dict(model, adapter, state)


@torch.no_grad()
def merge_adapter(model: nn.Module, name: str) -> int:
    """Fold adapter ``name`` into every LoraLinear's base weight (exact
    unmerge via snapshot).  Returns modules merged."""
    n = 0
    for _, module in iter_lora_modules(model):
        module.merge_adapter(name)
        n += 1
    return n


@torch.no_grad()
def unmerge_adapter(model: nn.Module) -> int:
    n = 0
    for _, module in iter_lora_modules(model):
        module.unmerge_adapter()
        n += 1
    return n
