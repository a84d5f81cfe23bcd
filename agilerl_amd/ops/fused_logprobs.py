"""Chunked fused-linear log-probabilities (the lm_head memory bottleneck).

Reference parity: ``agilerl/algorithms/core/llm_ops/fused_logprobs.py``
(``fused_linear_logprobs_chunked`` :135, exact-backward
``FusedLinearLogProbsFunction`` :201) — the reference leans on
torch.compile / Liger Triton; here the per-chunk GEMM runs on
rocBLAS/hipBLASLt and the softmax/gather/backward epilogues are the
hand-written CDNA4 kernels in ``lm_ops.hip``:

- fwd: per row-chunk ``logits = h W^T`` -> ``row_lse_gather`` (one-pass
  max+sumexp+target gather, fp32) -> the (chunk, V) logits buffer dies in
  registers/L2 and is never written back at full size.
- bwd: recompute the chunk's logits, ``row_softmax_bwd_`` overwrites them
  IN PLACE with ``(onehot - softmax) * dlp`` (no second (chunk, V)
  allocation), then two GEMMs accumulate grad_hidden and grad_weight.

Workspace is bounded to (chunk_rows, V) regardless of sequence length.
"""

from __future__ import annotations

from typing import Optional

import torch

from .backend import extension, use_hip

__all__ = ["fused_linear_logprobs", "resolve_chunk_rows"]


def resolve_chunk_rows(vocab_size: int, budget_bytes: int = 512 << 20) -> int:
    """Rows per chunk so the fp32 logits workspace stays under ``budget``."""
    rows = max(budget_bytes // (4 * max(vocab_size, 1)), 16)
    return int(min(rows, 8192))


def _chunk_fwd(h: torch.Tensor, weight: torch.Tensor, targets: torch.Tensor, inv_temp_t: float):
    ext = extension()
    logits = (h @ weight.t()).float()
    if use_hip(logits) and ext is not None:
        lp, lse = ext.row_lse_gather(logits.contiguous(), targets.contiguous(), 1.0 / inv_temp_t)
        return lp, lse
    scaled = logits * inv_temp_t
    lse = torch.logsumexp(scaled, dim=-1)
    lp = scaled.gather(1, targets.unsqueeze(1)).squeeze(1) - lse
    return lp, lse


class _FusedLinearLogProbs(torch.autograd.Function):
    @staticmethod
    def forward(ctx, hidden, weight, targets, temperature, chunk_rows):
        N = hidden.shape[0]
        inv_temp = 1.0 / temperature
        logprobs = torch.empty(N, device=hidden.device, dtype=torch.float32)
        lses = torch.empty(N, device=hidden.device, dtype=torch.float32)
        for start in range(0, N, chunk_rows):
            end = min(start + chunk_rows, N)
            lp, lse = _chunk_fwd(hidden[start:end], weight, targets[start:end], inv_temp)
            logprobs[start:end] = lp
            lses[start:end] = lse
        ctx.save_for_backward(hidden, weight, targets, lses)
        ctx.temperature = temperature
        ctx.chunk_rows = chunk_rows
        return logprobs

    @staticmethod
    def backward(ctx, grad_out):
        hidden, weight, targets, lses = ctx.saved_tensors
        inv_temp = 1.0 / ctx.temperature
        chunk_rows = ctx.chunk_rows
        ext = extension()
        N = hidden.shape[0]
        grad_out = grad_out.contiguous().float()
        need_h = ctx.needs_input_grad[0]
        need_w = ctx.needs_input_grad[1]
        grad_hidden = torch.zeros_like(hidden, dtype=torch.float32) if need_h else None
        grad_weight = torch.zeros_like(weight, dtype=torch.float32) if need_w else None
        for start in range(0, N, chunk_rows):
            end = min(start + chunk_rows, N)
            h_c = hidden[start:end]
            t_c = targets[start:end]
            g_c = grad_out[start:end]
            logits = (h_c @ weight.t()).float().contiguous()
            if use_hip(logits) and ext is not None:
                ext.row_softmax_bwd_(logits, t_c.contiguous(), lses[start:end].contiguous(),
                                     g_c, ctx.temperature)
                grad_logits = logits
            else:
                p = -torch.exp(logits * inv_temp - lses[start:end].unsqueeze(1))
                p[torch.arange(end - start, device=p.device), t_c] += 1.0
                grad_logits = p * g_c.unsqueeze(1) * inv_temp
            gl = grad_logits.to(weight.dtype)
            if need_h:
                grad_hidden[start:end] = (gl @ weight).float()
            if need_w:
                grad_weight += (gl.t() @ h_c).float()
        gh = grad_hidden.to(hidden.dtype) if need_h else None
        gw = grad_weight.to(weight.dtype) if need_w else None
        return gh, gw, None, None, None


def fused_linear_logprobs(
    hidden: torch.Tensor,
    weight: torch.Tensor,
    targets: torch.Tensor,
    temperature: float = 1.0,
    chunk_rows: Optional[int] = None,
) -> torch.Tensor:
    """log p(target | hidden) without materializing (N, V) logits.

    hidden: (N, H) or (B, T, H); weight: lm_head (V, H); targets: (N,) /
    (B, T) int64.  Returns fp32 logprobs with the leading shape of targets.
    """
    shape = targets.shape
    h2 = hidden.reshape(-1, hidden.shape[-1])
    t2 = targets.reshape(-1).long()
    if chunk_rows is None:
        chunk_rows = resolve_chunk_rows(weight.shape[0])
    out = _FusedLinearLogProbs.apply(h2, weight, t2, float(temperature), int(chunk_rows))
    return out.reshape(shape)
