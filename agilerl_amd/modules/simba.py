"""SimBa residual MLP encoder (Lee et al., "SimBa").

Reference parity: ``agilerl/modules/simba.py:13`` (EvolvableSimBa).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch
import torch.nn as nn

from .base import EvolvableModule, MutationType, mutation, preserve_parameters
from .components import SimbaResidualBlock

__all__ = ["EvolvableSimBa"]


class EvolvableSimBa(EvolvableModule):
    def __init__(
        self,
        num_inputs: int,
        num_outputs: int,
        hidden_size: int = 128,
        num_blocks: int = 2,
        min_blocks: int = 1,
        max_blocks: int = 4,
        min_mlp_nodes: int = 16,
        max_mlp_nodes: int = 500,
        scale_factor: int = 4,
        output_activation: Optional[str] = None,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        self.output_activation = output_activation
        self.num_inputs = int(num_inputs)
        self.num_outputs = int(num_outputs)
        self.hidden_size = int(hidden_size)
        self.num_blocks = int(num_blocks)
        self.min_blocks = min_blocks
        self.max_blocks = max_blocks
        self.min_mlp_nodes = min_mlp_nodes
        self.max_mlp_nodes = max_mlp_nodes
        self.scale_factor = scale_factor

        self.model = self._build().to(device)

    def _build(self) -> nn.Sequential:
        blocks = [nn.Linear(self.num_inputs, self.hidden_size)]
        for _ in range(self.num_blocks):
            blocks.append(SimbaResidualBlock(self.hidden_size, self.scale_factor))
        blocks.append(nn.LayerNorm(self.hidden_size))
        blocks.append(nn.Linear(self.hidden_size, self.num_outputs))
        if self.output_activation:
            from .components import get_activation

            blocks.append(get_activation(self.output_activation))
        return nn.Sequential(*blocks)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.model(x.float())

    @property
    def output_size(self) -> int:
        return self.num_outputs

    def reset_noise(self) -> None:
        pass

    def recreate_network(self) -> None:
        new_model = self._build().to(self.device)
        preserve_parameters(self.model, new_model)
        self.model = new_model

    @mutation(MutationType.LAYER)
    def add_block(self) -> dict:
        if self.num_blocks < self.max_blocks:
            self.num_blocks += 1
            self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def remove_block(self) -> dict:
        if self.num_blocks > self.min_blocks:
            self.num_blocks -= 1
            self.recreate_network()
        return {}

    @mutation(MutationType.NODE)
    def add_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([16, 32, 64]))
        if self.hidden_size + numb_new_nodes <= self.max_mlp_nodes:
            self.hidden_size += numb_new_nodes
            self.recreate_network()
        return {"numb_new_nodes": numb_new_nodes}

    @mutation(MutationType.NODE)
    def remove_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([16, 32, 64]))
        if self.hidden_size - numb_new_nodes >= self.min_mlp_nodes:
            self.hidden_size -= numb_new_nodes
            self.recreate_network()
        return {"numb_new_nodes": numb_new_nodes}
