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


def pool_log_ratio(
    log_ratio: torch.Tensor,
    mask: torch.Tensor,
    turn_ids: Optional[torch.Tensor],
    level: str,
) -> torch.Tensor:
    """Pool per-token log-ratios to the importance-sampling level.

    Reference parity: ``agilerl/algorithms/grpo.py:1848-1903``
    (``_log_importance_weights``).  "trajectory": length-normalized masked
    mean over the completion -> (B, 1) (GSPO's sequence-level ratio).
    "turn": the same geometric-mean pool restricted to each turn's tokens,
    scattered back to (B, T); degenerates to trajectory when ``turn_ids``
    is None.  Gradients flow through the pool (each token contributes
    mask/len to its pool's ratio).
    """
    mask_f = mask.float()
    if level == "trajectory" or turn_ids is None:
        return (log_ratio * mask_f).sum(1, keepdim=True) / mask_f.sum(
            1, keepdim=True
        ).clamp(min=1.0)
    K = max(int(turn_ids.max().item()) + 1, 1)
    safe = turn_ids.clamp(min=0).long()
    B = log_ratio.shape[0]
    num = torch.zeros(B, K, device=log_ratio.device, dtype=log_ratio.dtype)
    den = torch.zeros(B, K, device=log_ratio.device, dtype=log_ratio.dtype)
    num.scatter_add_(1, safe, log_ratio * mask_f)
    den.scatter_add_(1, safe, mask_f)
    pooled = num / den.clamp(min=1.0)
    return pooled.gather(1, safe)


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
    level: str = "token",
    turn_ids: Optional[torch.Tensor] = None,
    sampling_logp: Optional[torch.Tensor] = None,
    sampling_cap: float = 2.0,
    denom_tokens: Optional[float] = None,
    force_eager: bool = False,
) -> torch.Tensor:
    """Scalar policy loss over token tensors.

    ``loss_norm``: "token" divides by the total unmasked token count
    (or by ``denom_tokens`` when given — the accumulation-window
    normalizer, reference ``grpo.py:1619-1694``); "sequence" divides each
    sequence by its own length first (expects 2D (B, T) inputs).

    ``level``: importance-sampling granularity — "token" (GRPO),
    "turn" (per-turn pooled ratio via ``turn_ids``), "trajectory" (GSPO).

    ``sampling_logp``: behavior-policy (decode-engine) logprobs for the
    truncated importance-sampling correction — the analog of the
    reference's vLLM-IS correction (``grpo.py:2500-2510``): the policy
    term is reweighted by ``exp(old - sampling)`` clamped to
    ``sampling_cap``, detached.
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
        denom = (
            float(denom_tokens)
            if denom_tokens is not None
            else float(mask_f.sum().clamp(min=1.0))
        )

    ext = extension()
    fused_ok = level == "token" and sampling_logp is None
    if fused_ok and not force_eager and use_hip(logp) and ext is not None and logp.is_cuda:
        return _FusedGrpoLoss.apply(
            flat(logp), flat(old_logp),
            flat(ref_logp) if ref_logp is not None else None,
            flat(advantages), flat(mask_w),
            clip_lo, clip_hi, kl_coef, cispo, denom,
        )
    if fused_ok:
        loss_tok = _eager_token_loss(
            flat(logp.float()), flat(old_logp.float()),
            flat(ref_logp.float()) if ref_logp is not None else None,
            flat(advantages.float()), flat(mask_w), clip_lo, clip_hi, kl_coef, cispo,
        )
        return loss_tok.sum() / denom

    # pooled-IS / sampling-corrected path (eager; the heavy cost is the
    # logprob computation upstream, shared with the fused path)
    logp2 = logp.float()
    old2 = old_logp.float()
    log_ratio = logp2 - old2
    log_iw = pool_log_ratio(log_ratio, mask_f, turn_ids, level) if level != "token" else log_ratio
    ratio = log_iw.exp()
    adv = advantages.float()
    if cispo:
        loss = -(ratio.clamp(clip_lo, clip_hi).detach() * adv * logp2)
    else:
        loss = -torch.minimum(ratio * adv, ratio.clamp(clip_lo, clip_hi) * adv)
    if sampling_logp is not None:
        with torch.no_grad():
            is_ratio = ((old2 - sampling_logp.float()) * mask_f).exp().clamp(max=sampling_cap)
        loss = loss * is_ratio
    if ref_logp is not None and kl_coef:
        d = ref_logp.float() - logp2
        loss = loss + kl_coef * (d.exp() - d - 1)
    return (loss * mask_w).sum() / denom
