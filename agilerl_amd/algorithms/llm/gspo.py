"""GSPO — trajectory-level importance sampling GRPO variant.

Reference parity: ``agilerl/algorithms/gspo.py:15`` (``_gspo_loss``,
grpo.py:2028): the per-token log-ratio is pooled by masked mean per
sequence, giving one IS weight s_i = exp(mean_t log ratio) per
trajectory; the clip/min surrogate applies at the sequence level.  The
heavy lifting (fused logprob fwd/bwd) is unchanged; the pooled surrogate
itself is O(B*T) elementwise.
"""

from __future__ import annotations

from .grpo import GRPO

__all__ = ["GSPO"]


class GSPO(GRPO):
    """GRPO with ``importance_sampling_level`` defaulting to "trajectory":
    the surrogate runs through the shared pooled-IS path in
    ``ops.grpo_loss.grpo_policy_loss`` (``pool_log_ratio``), exactly the
    reference's ``_gspo_loss`` -> ``_compute_policy_loss(level=
    "trajectory")`` delegation."""

    SEQUENCE_LEVEL_IS = True
