"""State-value network V(s).

Reference parity: ``agilerl/networks/value_networks.py:16`` (ValueNetwork).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from ..spaces import Space
from .base import EvolvableNetwork

__all__ = ["ValueNetwork"]


class ValueNetwork(EvolvableNetwork):
    def __init__(
        self,
        observation_space: Space,
        encoder_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        latent_dim: int = 64,
        device: str = "cpu",
        **net_kwargs,
    ):
        super().__init__(
            observation_space,
            num_outputs=1,
            encoder_config=encoder_config,
            head_config=head_config,
            latent_dim=latent_dim,
            device=device,
            **net_kwargs,
        )
