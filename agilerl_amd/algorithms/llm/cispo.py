"""CISPO — clamped-IS x logp objective (GRPO variant).

Reference parity: ``agilerl/algorithms/cispo.py:15`` (objective at
grpo.py:2051): the clamped ratio is treated as a constant weight and the
gradient flows through ``logp`` directly (REINFORCE-with-clamped-weight).
The fused HIP kernel implements this via the ``cispo`` flag.
"""

from .grpo import GRPO

__all__ = ["CISPO"]


class CISPO(GRPO):
    CISPO = True
