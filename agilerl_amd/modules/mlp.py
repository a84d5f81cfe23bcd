"""Evolvable multi-layer perceptron.

Reference parity: ``agilerl/modules/mlp.py`` (EvolvableMLP :13, mutation
methods ``add_layer`` :231 / ``remove_layer`` :245 / ``add_node`` :258 /
``remove_node`` :288 / ``recreate_network`` :317).  New implementation:
layers are rebuilt from the live ``hidden_size`` list with parameter slices
preserved; optional NoisyLinear layers route through the fused HIP noisy-GEMM
on GPU (``agilerl_amd/ops``).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch
import torch.nn as nn

from .base import EvolvableModule, MutationType, mutation, preserve_parameters
from .components import NoisyLinear, get_activation

__all__ = ["EvolvableMLP", "create_mlp"]


def create_mlp(
    input_size: int,
    output_size: int,
    hidden_size: List[int],
    activation: str = "ReLU",
    output_activation: Optional[str] = None,
    noisy: bool = False,
    noise_std: float = 0.5,
    layer_norm: bool = False,
    output_layernorm: bool = False,
    init_layers: bool = True,
    output_vanish: bool = False,
) -> nn.Sequential:
    """Plain (non-evolvable) MLP builder used for network heads."""

    def linear(in_f: int, out_f: int) -> nn.Module:
        if noisy:
            return NoisyLinear(in_f, out_f, std_init=noise_std)
        layer = nn.Linear(in_f, out_f)
        if init_layers:
            nn.init.orthogonal_(layer.weight, gain=float(np.sqrt(2)))
            nn.init.zeros_(layer.bias)
        return layer

    layers: List[nn.Module] = []
    sizes = [input_size] + list(hidden_size)
    for i in range(len(sizes) - 1):
        layers.append(linear(sizes[i], sizes[i + 1]))
        if layer_norm:
            layers.append(nn.LayerNorm(sizes[i + 1]))
        layers.append(get_activation(activation))
    out_layer = linear(sizes[-1], output_size)
    if output_vanish and not noisy:
        with torch.no_grad():
            out_layer.weight.mul_(0.1)
            if out_layer.bias is not None:
                out_layer.bias.mul_(0.1)
    layers.append(out_layer)
    if output_layernorm:
        layers.append(nn.LayerNorm(output_size))
    if output_activation is not None:
        layers.append(get_activation(output_activation))
    return nn.Sequential(*layers)


class EvolvableMLP(EvolvableModule):
    """MLP whose depth/width evolve under population-based training."""

    def __init__(
        self,
        num_inputs: int,
        num_outputs: int,
        hidden_size: Optional[List[int]] = None,
        activation: str = "ReLU",
        output_activation: Optional[str] = None,
        min_hidden_layers: int = 1,
        max_hidden_layers: int = 3,
        min_mlp_nodes: int = 16,
        max_mlp_nodes: int = 500,
        layer_norm: bool = False,
        output_layernorm: bool = False,
        noisy: bool = False,
        noise_std: float = 0.5,
        init_layers: bool = True,
        output_vanish: bool = False,
        new_gelu: bool = False,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        if new_gelu:
            # reference mlp.py new_gelu: tanh-approx GELU activations
            activation = "NewGELU"
        self.new_gelu = bool(new_gelu)
        self.num_inputs = int(num_inputs)
        self.num_outputs = int(num_outputs)
        self.hidden_size = list(hidden_size) if hidden_size is not None else [64, 64]
        self.activation = activation
        self.output_activation = output_activation
        self.min_hidden_layers = min_hidden_layers
        self.max_hidden_layers = max_hidden_layers
        self.min_mlp_nodes = min_mlp_nodes
        self.max_mlp_nodes = max_mlp_nodes
        self.layer_norm = layer_norm
        self.output_layernorm = output_layernorm
        self.noisy = noisy
        self.noise_std = noise_std
        self.init_layers = init_layers
        # reference-parity knob (agilerl mlp config): scale the final
        # layer's weights/bias by 0.1 so initial outputs start near zero
        self.output_vanish = output_vanish

        self.model = self._build().to(device)

    # ------------------------------------------------------------------
    def _build(self) -> nn.Sequential:
        return create_mlp(
            self.num_inputs,
            self.num_outputs,
            self.hidden_size,
            activation=self.activation,
            output_activation=self.output_activation,
            noisy=self.noisy,
            noise_std=self.noise_std,
            layer_norm=self.layer_norm,
            output_layernorm=self.output_layernorm,
            init_layers=self.init_layers,
            output_vanish=self.output_vanish,
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dtype != torch.float32 and not x.is_floating_point():
            x = x.float()
        return self.model(x.reshape(x.shape[0], -1) if x.dim() > 2 else x)

    @property
    def output_size(self) -> int:
        return self.num_outputs

    def reset_noise(self) -> None:
        for m in self.model.modules():
            if isinstance(m, NoisyLinear):
                m.reset_noise()

    # ------------------------------------------------------------------
    # Mutations
    # ------------------------------------------------------------------
    def recreate_network(self) -> None:
        new_model = self._build().to(self.device)
        preserve_parameters(self.model, new_model)
        self.model = new_model

    @mutation(MutationType.LAYER)
    def add_layer(self, hidden_layer: Optional[int] = None, numb_new_nodes: Optional[int] = None) -> dict:
        # fallback choices are accepted so a replay on a sibling network takes
        # the identical path even when the depth bound triggers the fallback
        if len(self.hidden_size) >= self.max_hidden_layers:
            return self.add_node(hidden_layer, numb_new_nodes)
        self.hidden_size.append(self.hidden_size[-1])
        self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def remove_layer(self, hidden_layer: Optional[int] = None, numb_new_nodes: Optional[int] = None) -> dict:
        if len(self.hidden_size) <= self.min_hidden_layers:
            return self.add_node(hidden_layer, numb_new_nodes)
        self.hidden_size.pop()
        self.recreate_network()
        return {}

    @mutation(MutationType.NODE)
    def add_node(self, hidden_layer: Optional[int] = None, numb_new_nodes: Optional[int] = None) -> dict:
        if hidden_layer is None:
            hidden_layer = int(np.random.randint(0, len(self.hidden_size)))
        else:
            hidden_layer = min(hidden_layer, len(self.hidden_size) - 1)
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([16, 32, 64]))
        if self.hidden_size[hidden_layer] + numb_new_nodes <= self.max_mlp_nodes:
            self.hidden_size[hidden_layer] += numb_new_nodes
            self.recreate_network()
        return {"hidden_layer": hidden_layer, "numb_new_nodes": numb_new_nodes}

    @mutation(MutationType.NODE)
    def remove_node(self, hidden_layer: Optional[int] = None, numb_new_nodes: Optional[int] = None) -> dict:
        if hidden_layer is None:
            hidden_layer = int(np.random.randint(0, len(self.hidden_size)))
        else:
            hidden_layer = min(hidden_layer, len(self.hidden_size) - 1)
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([16, 32, 64]))
        if self.hidden_size[hidden_layer] - numb_new_nodes >= self.min_mlp_nodes:
            self.hidden_size[hidden_layer] -= numb_new_nodes
            self.recreate_network()
        return {"hidden_layer": hidden_layer, "numb_new_nodes": numb_new_nodes}

    @mutation(MutationType.ACTIVATION)
    def change_activation(self, activation: Optional[str] = None, output: bool = False) -> dict:
        if activation is None:
            activation = str(np.random.choice(["ReLU", "ELU", "GELU"]))
        self.activation = activation
        if output:
            self.output_activation = activation
        self.recreate_network()
        return {"activation": activation, "output": output}
