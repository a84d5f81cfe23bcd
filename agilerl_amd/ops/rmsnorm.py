"""Fused RMSNorm (autograd wrapper over the CDNA4 kernels)."""

from __future__ import annotations

import torch

from .backend import extension, use_hip

__all__ = ["rms_norm", "HipRMSNorm"]


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = extension()
        y, inv_rms = ext.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, inv_rms)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = extension()
        x, weight, inv_rms = ctx.saved_tensors
        dx, dw = ext.rmsnorm_bwd(x, weight, dy, inv_rms)
        return dx, dw.to(weight.dtype), None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """y = x * weight / sqrt(mean(x^2, -1) + eps).

    GPU bf16 inputs go through the fused HIP kernels (one pass each way);
    everything else uses the eager fp32-accumulated reference."""
    ext = extension()
    if (
        use_hip(x)
        and ext is not None
        and x.dtype == torch.bfloat16
        and x.shape[-1] % 8 == 0
    ):
        return _RMSNormFn.apply(x, weight.to(torch.bfloat16), float(eps))
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (y * weight.float()).to(x.dtype)


class HipRMSNorm(torch.nn.Module):
    """Drop-in replacement for transformers' LlamaRMSNorm."""

    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rms_norm(x, self.weight, self.variance_epsilon)

    def extra_repr(self) -> str:
        return f"{self.weight.shape[0]}, eps={self.variance_epsilon}"
