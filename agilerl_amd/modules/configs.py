"""Net-config dataclasses for the evolvable architectures.

Reference parity: ``agilerl/modules/configs.py``.  These are plain
dataclasses usable both directly and as the ``net_config`` dict payloads in
training manifests (see ``agilerl_amd/models/networks.py``).
"""

from __future__ import annotations

from dataclasses import asdict, dataclass, field
from typing import Any, Dict, List, Optional

__all__ = [
    "MlpNetConfig",
    "CnnNetConfig",
    "LstmNetConfig",
    "SimbaNetConfig",
    "MultiInputNetConfig",
    "NetConfig",
]


@dataclass
class NetConfig:
    def to_dict(self) -> Dict[str, Any]:
        return asdict(self)


@dataclass
class MlpNetConfig(NetConfig):
    hidden_size: List[int] = field(default_factory=lambda: [64, 64])
    activation: str = "ReLU"
    output_activation: Optional[str] = None
    min_hidden_layers: int = 1
    max_hidden_layers: int = 3
    min_mlp_nodes: int = 16
    max_mlp_nodes: int = 500
    layer_norm: bool = False
    output_layernorm: bool = False
    noisy: bool = False
    noise_std: float = 0.5
    init_layers: bool = True


@dataclass
class CnnNetConfig(NetConfig):
    channel_size: List[int] = field(default_factory=lambda: [32, 32])
    kernel_size: List[int] = field(default_factory=lambda: [3, 3])
    stride_size: List[int] = field(default_factory=lambda: [1, 1])
    activation: str = "ReLU"
    min_hidden_layers: int = 1
    max_hidden_layers: int = 6
    min_channel_size: int = 16
    max_channel_size: int = 256
    layer_norm: bool = False


@dataclass
class LstmNetConfig(NetConfig):
    hidden_state_size: int = 64
    num_layers: int = 1
    min_hidden_state_size: int = 16
    max_hidden_state_size: int = 500
    min_layers: int = 1
    max_layers: int = 3


@dataclass
class SimbaNetConfig(NetConfig):
    hidden_size: int = 128
    num_blocks: int = 2
    min_blocks: int = 1
    max_blocks: int = 4
    min_mlp_nodes: int = 16
    max_mlp_nodes: int = 500
    scale_factor: int = 4


@dataclass
class MultiInputNetConfig(NetConfig):
    latent_dim: int = 64
    mlp_config: Optional[Dict[str, Any]] = None
    cnn_config: Optional[Dict[str, Any]] = None
    vector_space_mlp: bool = True
