"""Evolvable convolutional encoder.

Reference parity: ``agilerl/modules/cnn.py:245`` (EvolvableCNN, mutable
kernel/channel sizes :58).  Produces a flat feature vector from (C, H, W)
image observations; conv stack is mutable in depth, channel width and
kernel size, with parameter preservation across rebuilds.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np
import torch
import torch.nn as nn

from .base import EvolvableModule, MutationType, mutation, preserve_parameters
from .components import get_activation

__all__ = ["EvolvableCNN"]


class EvolvableCNN(EvolvableModule):
    def __init__(
        self,
        input_shape: Tuple[int, int, int],
        num_outputs: int,
        channel_size: Optional[List[int]] = None,
        kernel_size: Optional[List[int]] = None,
        stride_size: Optional[List[int]] = None,
        activation: str = "ReLU",
        output_activation: Optional[str] = None,
        min_hidden_layers: int = 1,
        max_hidden_layers: int = 6,
        min_channel_size: int = 16,
        max_channel_size: int = 256,
        layer_norm: bool = False,
        block_type: str = "Conv2d",
        sample_input: Optional[torch.Tensor] = None,
        init_layers: bool = True,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        # reference cnn.py:15 BlockType: Conv1d/2d/3d selected by the obs
        # rank; sample_input overrides the flat-dim inference for exotic
        # shapes; init_layers toggles orthogonal-style re-init (our layers
        # use PyTorch defaults already, kept for API compat)
        if block_type not in ("Conv1d", "Conv2d", "Conv3d"):
            raise ValueError("block_type must be Conv1d | Conv2d | Conv3d")
        self.block_type = block_type
        self.sample_input = sample_input
        self.init_layers = bool(init_layers)
        self.input_shape = tuple(input_shape)
        self.num_outputs = int(num_outputs)
        self.channel_size = list(channel_size) if channel_size is not None else [32, 32]
        self.kernel_size = list(kernel_size) if kernel_size is not None else [3] * len(self.channel_size)
        self.stride_size = list(stride_size) if stride_size is not None else [1] * len(self.channel_size)
        self.activation = activation
        self.output_activation = output_activation
        self.min_hidden_layers = min_hidden_layers
        self.max_hidden_layers = max_hidden_layers
        self.min_channel_size = min_channel_size
        self.max_channel_size = max_channel_size
        self.layer_norm = layer_norm

        self.model = self._build().to(device)

    @property
    def _conv_cls(self):
        return {"Conv1d": nn.Conv1d, "Conv2d": nn.Conv2d, "Conv3d": nn.Conv3d}[self.block_type]

    def _conv_output_dim(self, conv: nn.Sequential) -> int:
        with torch.no_grad():
            if self.sample_input is not None:
                dummy = self.sample_input.to("cpu").float()
                if dummy.dim() == len(self.input_shape):
                    dummy = dummy.unsqueeze(0)
            else:
                dummy = torch.zeros(1, *self.input_shape)
            return int(np.prod(conv(dummy).shape[1:]))

    def _build(self) -> nn.Sequential:
        layers: List[nn.Module] = []
        in_ch = self.input_shape[0]
        for out_ch, k, s in zip(self.channel_size, self.kernel_size, self.stride_size):
            layers.append(self._conv_cls(in_ch, out_ch, k, s, padding=k // 2))
            if self.layer_norm:
                layers.append(nn.GroupNorm(1, out_ch))
            layers.append(get_activation(self.activation))
            in_ch = out_ch
        conv = nn.Sequential(*layers)
        flat = self._conv_output_dim(conv)
        head: List[nn.Module] = [conv, nn.Flatten(), nn.Linear(flat, self.num_outputs)]
        if self.output_activation is not None:
            head.append(get_activation(self.output_activation))
        return nn.Sequential(*head)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == len(self.input_shape):
            x = x.unsqueeze(0)
        if not x.is_floating_point():
            x = x.float() / 255.0
        return self.model(x)

    @property
    def output_size(self) -> int:
        return self.num_outputs

    def reset_noise(self) -> None:
        pass

    # ------------------------------------------------------------------
    def recreate_network(self) -> None:
        new_model = self._build().to(self.device)
        preserve_parameters(self.model, new_model)
        self.model = new_model

    @mutation(MutationType.LAYER)
    def add_layer(self, hidden_layer: Optional[int] = None, numb_new_channels: Optional[int] = None) -> dict:
        if len(self.channel_size) >= self.max_hidden_layers:
            return self.add_channel(hidden_layer, numb_new_channels)
        self.channel_size.append(self.channel_size[-1])
        self.kernel_size.append(3)
        self.stride_size.append(1)
        self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def remove_layer(self, hidden_layer: Optional[int] = None, numb_new_channels: Optional[int] = None) -> dict:
        if len(self.channel_size) <= self.min_hidden_layers:
            return self.add_channel(hidden_layer, numb_new_channels)
        self.channel_size.pop()
        self.kernel_size.pop()
        self.stride_size.pop()
        self.recreate_network()
        return {}

    @mutation(MutationType.NODE)
    def add_channel(self, hidden_layer: Optional[int] = None, numb_new_channels: Optional[int] = None) -> dict:
        if hidden_layer is None:
            hidden_layer = int(np.random.randint(0, len(self.channel_size)))
        else:
            hidden_layer = min(hidden_layer, len(self.channel_size) - 1)
        if numb_new_channels is None:
            numb_new_channels = int(np.random.choice([8, 16, 32]))
        if self.channel_size[hidden_layer] + numb_new_channels <= self.max_channel_size:
            self.channel_size[hidden_layer] += numb_new_channels
            self.recreate_network()
        return {"hidden_layer": hidden_layer, "numb_new_channels": numb_new_channels}

    @mutation(MutationType.NODE)
    def remove_channel(self, hidden_layer: Optional[int] = None, numb_new_channels: Optional[int] = None) -> dict:
        if hidden_layer is None:
            hidden_layer = int(np.random.randint(0, len(self.channel_size)))
        else:
            hidden_layer = min(hidden_layer, len(self.channel_size) - 1)
        if numb_new_channels is None:
            numb_new_channels = int(np.random.choice([8, 16, 32]))
        if self.channel_size[hidden_layer] - numb_new_channels >= self.min_channel_size:
            self.channel_size[hidden_layer] -= numb_new_channels
            self.recreate_network()
        return {"hidden_layer": hidden_layer, "numb_new_channels": numb_new_channels}

    @mutation(MutationType.NODE)
    def change_kernel(self, hidden_layer: Optional[int] = None, kernel_size: Optional[int] = None) -> dict:
        if len(self.channel_size) == 0:
            return {}
        if hidden_layer is None:
            hidden_layer = int(np.random.randint(0, len(self.kernel_size)))
        else:
            hidden_layer = min(hidden_layer, len(self.kernel_size) - 1)
        if kernel_size is None:
            kernel_size = int(np.random.choice([3, 5, 7]))
        self.kernel_size[hidden_layer] = kernel_size
        self.recreate_network()
        return {"hidden_layer": hidden_layer, "kernel_size": kernel_size}

    @mutation(MutationType.ACTIVATION)
    def change_activation(self, activation: Optional[str] = None, output: bool = False) -> dict:
        if activation is None:
            activation = str(np.random.choice(["ReLU", "ELU", "GELU"]))
        self.activation = activation
        if output:
            self.output_activation = activation
        self.recreate_network()
        return {"activation": activation, "output": output}
