"""Fused SwiGLU activation (autograd wrapper over the CDNA4 kernels).

Analog of the reference's Liger SwiGLU patch (SURVEY §2.9.13,
architectures/nemotron_h/liger.py): ``silu(gate) * up`` as one kernel
each way instead of three elementwise passes.  The gate/up/down GEMMs
stay on hipBLASLt.  Opt-in for HF Llama via
``architectures.llama_patches.patch_llama_swiglu``.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from .backend import extension, use_hip

__all__ = ["swiglu"]


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        ext = extension()
        out = ext.swiglu_fwd(gate, up)
        ctx.save_for_backward(gate, up)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = extension()
        gate, up = ctx.saved_tensors
        dg, du = ext.swiglu_bwd(gate, up, dout)
        return dg, du


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up — fused on GPU bf16, eager elsewhere."""
    ext = extension()
    if (
        use_hip(gate)
        and ext is not None
        and gate.dtype == torch.bfloat16
        and up.dtype == torch.bfloat16
    ):
        return _SwiGLUFn.apply(gate, up)
    return F.silu(gate) * up
