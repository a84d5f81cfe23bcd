"""Model-family kernel patches.

Reference parity: ``agilerl/architectures/nemotron_h/liger.py:181``
(``apply_liger_kernel_to_nemotron_h`` swaps model internals for fused
Triton kernels).  This framework targets Llama-family models: the HF
eager RMSNorm chain (cast/pow/mean/rsqrt/mul/mul = several HBM round
trips per call) is replaced by the one-pass CDNA4 kernels in
``ops/csrc/norm_ops.hip``.
"""

from __future__ import annotations

import torch.nn as nn

from ..ops.rmsnorm import HipRMSNorm

__all__ = ["apply_hip_kernels_to_llama", "patch_llama_swiglu"]


def apply_hip_kernels_to_llama(model: nn.Module) -> int:
    """Swap every *RMSNorm module for the fused HIP implementation
    (weights carried over).  Returns the number of modules patched."""
    patched = 0
    for parent_name, parent in list(model.named_modules()):
        for child_name, child in list(parent.named_children()):
            if type(child).__name__.endswith("RMSNorm") and hasattr(child, "weight"):
                eps = getattr(child, "variance_epsilon", getattr(child, "eps", 1e-6))
                new = HipRMSNorm(child.weight.shape[0], eps=eps)
                new = new.to(child.weight.device, child.weight.dtype)
                with __import__("torch").no_grad():
                    new.weight.copy_(child.weight)
                new.weight.requires_grad = child.weight.requires_grad
                setattr(parent, child_name, new)
                patched += 1
    return patched


def patch_llama_swiglu(model: nn.Module) -> int:
    """Swap each Llama MLP's ``act_fn(gate) * up`` for the fused SwiGLU
    kernel (ops/csrc/act_ops.hip).  Opt-in: call after model creation.
    Returns the number of MLP modules patched."""
    from ..ops.swiglu import swiglu

    patched = 0
    for module in model.modules():
        if (
            type(module).__name__.endswith("MLP")
            and hasattr(module, "gate_proj")
            and hasattr(module, "up_proj")
            and hasattr(module, "down_proj")
        ):
            def fused_forward(x, _m=module):
                return _m.down_proj(swiglu(_m.gate_proj(x), _m.up_proj(x)))

            module.forward = fused_forward
            patched += 1
    return patched
