"""Fused token-masked GRPO / GSPO / CISPO surrogate loss.

Reference parity: ``agilerl/algorithms/grpo.py:1904-2074``
(``_compute_policy_loss``: masked fill, k3 KL, clip/min surrogate, masked
reductions) and the Liger fused path (``llm_ops/fused_loss.py``).  On GPU
the whole token loss + its analytic d(loss)/d(logp) come from ONE HIP
kernel pass (``lm_ops.hip::grpo_token_loss``); backward into the policy
logprobs is a single multiply.
"""

from __future__ import annotations

from typing import Optional

import torch

from .backend import extension, use_hip

__all__ = ["grpo_policy_loss"]


def _eager_token_loss(logp, old_logp, ref_logp, adv, mask, clip_lo, clip_hi, kl_coef, cispo):
    ratio = (logp - old_logp).exp()
    if cispo:
        w = ratio.clamp(clip_lo, clip_hi).detach()
        loss = -w * adv * logp
    else:
        loss = -torch.minimum(ratio * adv, ratio.clamp(clip_lo, clip_hi) * adv)
    if ref_logp is not None and kl_coef:
        d = ref_logp - logp
        loss = loss + kl_coef * (d.exp() - d - 1)
    return loss * mask


class _FusedGrpoLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logp, old_logp, ref_logp, adv, mask, clip_lo, clip_hi, kl_coef, cispo, denom):
        ext = extension()
        loss_tok, dlogp = ext.grpo_token_loss(
            logp.contiguous().float(),
            old_logp.contiguous().float(),
            ref_logp.contiguous().float() if ref_logp is not None else None,
            adv.contiguous().float(),
            mask.contiguous().float(),
            float(clip_lo), float(clip_hi), float(kl_coef), bool(cispo),
        )
        ctx.save_for_backward(dlogp)
        ctx.denom = denom
        return loss_tok.sum() / denom

    @staticmethod
    def backward(ctx, grad_out):
        (dlogp,) = ctx.saved_tensors
        g = dlogp * (grad_out / ctx.denom)
        return g, None, None, None, None, None, None, None, None, None


def grpo_policy_loss(
    logp: torch.Tensor,
    old_logp: torch.Tensor,
    advantages: torch.Tensor,
    mask: torch.Tensor,
    ref_logp: Optional[torch.Tensor] = None,
    clip_lo: float = 0.8,
    clip_hi: float = 1.2,
    kl_coef: float = 0.0,
    cispo: bool = False,
    loss_norm: str = "token",
) -> torch.Tensor:
    """Scalar policy loss over flattened token tensors (any shape).

    ``loss_norm``: "token" divides by the total unmasked token count;
    "sequence" divides each sequence by its own length first (expects 2D
    (B, T) inputs).
    """
    shape = logp.shape
    flat = lambda t: t.reshape(-1)
    mask_f = mask.float()
    if loss_norm == "sequence" and logp.dim() == 2:
        seq_len = mask_f.sum(dim=1, keepdim=True).clamp(min=1.0)
        # fold per-sequence normalization into the mask weights
        mask_w = mask_f / seq_len
        denom = float(shape[0])
    else:
        mask_w = mask_f
        denom = float(mask_f.sum().clamp(min=1.0))

    ext = extension()
    if use_hip(logp) and ext is not None and logp.is_cuda:
        return _FusedGrpoLoss.apply(
            flat(logp), flat(old_logp),
            flat(ref_logp) if ref_logp is not None else None,
            flat(advantages), flat(mask_w),
            clip_lo, clip_hi, kl_coef, cispo, denom,
        )
    loss_tok = _eager_token_loss(
        flat(logp.float()), flat(old_logp.float()),
        flat(ref_logp.float()) if ref_logp is not None else None,
        flat(advantages.float()), flat(mask_w), clip_lo, clip_hi, kl_coef, cispo,
    )
    return loss_tok.sum() / denom
