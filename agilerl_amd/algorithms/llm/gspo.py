"""GSPO — trajectory-level importance sampling GRPO variant.

Reference parity: ``agilerl/algorithms/gspo.py:15`` (``_gspo_loss``,
grpo.py:2028): the per-token log-ratio is pooled by masked mean per
sequence, giving one IS weight s_i = exp(mean_t log ratio) per
trajectory; the clip/min surrogate applies at the sequence level.  The
heavy lifting (fused logprob fwd/bwd) is unchanged; the pooled surrogate
itself is O(B*T) elementwise.
"""

from __future__ import annotations

import torch

from ... import ops
from .grpo import GRPO

__all__ = ["GSPO"]


class GSPO(GRPO):
    SEQUENCE_LEVEL_IS = True

    def _policy_loss(self, logp, old_logp, adv_tok, mask, ref_logp, clip_lo, clip_hi):
        mask_f = mask.float()
        seq_len = mask_f.sum(dim=1).clamp(min=1.0)
        log_ratio_seq = ((logp - old_logp) * mask_f).sum(dim=1) / seq_len  # (B,)
        s = log_ratio_seq.exp()
        adv_seq = (adv_tok * mask_f).sum(dim=1) / seq_len  # constant per seq
        surr = torch.minimum(s * adv_seq, s.clamp(clip_lo, clip_hi) * adv_seq)
        loss = -surr.mean()
        if ref_logp is not None and self.beta:
            d = ref_logp - logp
            loss = loss + self.beta * ops.masked_mean(d.exp() - d - 1, mask_f)
        return loss
