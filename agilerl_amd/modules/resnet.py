"""Evolvable ResNet encoder (conv residual stacks).

Reference parity: ``agilerl/modules/resnet.py:15`` (EvolvableResNet).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch
import torch.nn as nn

from .base import EvolvableModule, MutationType, mutation, preserve_parameters
from .components import ResidualBlock

__all__ = ["EvolvableResNet"]


class EvolvableResNet(EvolvableModule):
    def __init__(
        self,
        input_shape: Tuple[int, int, int],
        num_outputs: int,
        channel_size: int = 32,
        num_blocks: int = 2,
        min_blocks: int = 1,
        max_blocks: int = 6,
        min_channel_size: int = 16,
        max_channel_size: int = 256,
        kernel_size: int = 3,
        stride_size: int = 1,
        scale_factor: int = 4,
        output_activation: Optional[str] = None,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        self.kernel_size = int(kernel_size)
        self.stride_size = int(stride_size)
        self.scale_factor = int(scale_factor)
        self.output_activation = output_activation
        self.input_shape = tuple(input_shape)
        self.num_outputs = int(num_outputs)
        self.channel_size = int(channel_size)
        self.num_blocks = int(num_blocks)
        self.min_blocks = min_blocks
        self.max_blocks = max_blocks
        self.min_channel_size = min_channel_size
        self.max_channel_size = max_channel_size
        self.model = self._build().to(device)

    def _build(self) -> nn.Sequential:
        layers = [
            nn.Conv2d(self.input_shape[0], self.channel_size, 3, 2, 1),
            nn.BatchNorm2d(self.channel_size),
            nn.ReLU(),
        ]
        for _ in range(self.num_blocks):
            layers.append(ResidualBlock(self.channel_size))
        layers.append(nn.AdaptiveAvgPool2d((4, 4)))
        layers.append(nn.Flatten())
        layers.append(nn.Linear(self.channel_size * 16, self.num_outputs))
        if self.output_activation:
            from .components import get_activation

            layers.append(get_activation(self.output_activation))
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 3:
            x = x.unsqueeze(0)
        if not x.is_floating_point():
            x = x.float() / 255.0
        return self.model(x)

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
    def add_channel(self, numb_new_channels: Optional[int] = None) -> dict:
        if numb_new_channels is None:
            numb_new_channels = int(np.random.choice([8, 16, 32]))
        if self.channel_size + numb_new_channels <= self.max_channel_size:
            self.channel_size += numb_new_channels
            self.recreate_network()
        return {"numb_new_channels": numb_new_channels}

    @mutation(MutationType.NODE)
    def remove_channel(self, numb_new_channels: Optional[int] = None) -> dict:
        if numb_new_channels is None:
            numb_new_channels = int(np.random.choice([8, 16, 32]))
        if self.channel_size - numb_new_channels >= self.min_channel_size:
            self.channel_size -= numb_new_channels
            self.recreate_network()
        return {"numb_new_channels": numb_new_channels}
