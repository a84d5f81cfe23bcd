"""Evolvable LSTM encoder (recurrent policies / BPTT PPO).

Reference parity: ``agilerl/modules/lstm.py:14`` (EvolvableLSTM).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch
import torch.nn as nn

from .base import EvolvableModule, MutationType, mutation, preserve_parameters

__all__ = ["EvolvableLSTM"]


class EvolvableLSTM(EvolvableModule):
    def __init__(
        self,
        input_size: int,
        num_outputs: int,
        hidden_state_size: int = 64,
        num_layers: int = 1,
        min_hidden_state_size: int = 16,
        max_hidden_state_size: int = 500,
        min_layers: int = 1,
        max_layers: int = 3,
        dropout: float = 0.0,
        output_activation: Optional[str] = None,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        self.dropout = float(dropout)
        self.output_activation = output_activation
        self.input_size = int(input_size)
        self.num_outputs = int(num_outputs)
        self.hidden_state_size = int(hidden_state_size)
        self.num_layers = int(num_layers)
        self.min_hidden_state_size = min_hidden_state_size
        self.max_hidden_state_size = max_hidden_state_size
        self.min_layers = min_layers
        self.max_layers = max_layers

        self.lstm = self._build_lstm().to(device)
        self.proj = nn.Linear(self.hidden_state_size, self.num_outputs).to(device)
        from .components import get_activation

        self.out_act = get_activation(output_activation).to(device)

    def _build_lstm(self) -> nn.LSTM:
        return nn.LSTM(
            self.input_size, self.hidden_state_size, self.num_layers,
            batch_first=True,
            dropout=self.dropout if self.num_layers > 1 else 0.0,
        )

    def forward(
        self,
        x: torch.Tensor,
        hidden: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
    ) -> torch.Tensor:
        """x: (B, T, F) or (B, F). Returns features of the last timestep."""
        single_step = x.dim() == 2
        if single_step:
            x = x.unsqueeze(1)
        out, self._hidden_out = self.lstm(x.float(), hidden)
        feats = self.proj(out[:, -1])
        return feats

    def forward_sequence(
        self, x: torch.Tensor, hidden: Optional[Tuple[torch.Tensor, torch.Tensor]] = None
    ) -> Tuple[torch.Tensor, Tuple[torch.Tensor, torch.Tensor]]:
        out, hidden = self.lstm(x.float(), hidden)
        return self.out_act(self.proj(out)), hidden

    def initial_hidden(self, batch_size: int) -> Tuple[torch.Tensor, torch.Tensor]:
        h = torch.zeros(self.num_layers, batch_size, self.hidden_state_size, device=self.device)
        return h, h.clone()

    def step(
        self, x: torch.Tensor, hidden: Optional[Tuple[torch.Tensor, torch.Tensor]] = None
    ) -> Tuple[torch.Tensor, Tuple[torch.Tensor, torch.Tensor]]:
        """Single-timestep recurrence: x (B, F) -> (features (B, out), hidden)."""
        if hidden is None:
            hidden = self.initial_hidden(x.shape[0])
        out, new_hidden = self.lstm(x.float().unsqueeze(1), hidden)
        return self.out_act(self.proj(out[:, -1])), new_hidden

    @property
    def output_size(self) -> int:
        return self.num_outputs

    def reset_noise(self) -> None:
        pass

    # ------------------------------------------------------------------
    def recreate_network(self) -> None:
        new_lstm = self._build_lstm().to(self.device)
        preserve_parameters(self.lstm, new_lstm)
        self.lstm = new_lstm
        new_proj = nn.Linear(self.hidden_state_size, self.num_outputs).to(self.device)
        preserve_parameters(self.proj, new_proj)
        self.proj = new_proj

    @mutation(MutationType.LAYER)
    def add_layer(self) -> dict:
        if self.num_layers < self.max_layers:
            self.num_layers += 1
            self.recreate_network()
        return {}

    @mutation(MutationType.LAYER)
    def remove_layer(self) -> dict:
        if self.num_layers > self.min_layers:
            self.num_layers -= 1
            self.recreate_network()
        return {}

    @mutation(MutationType.NODE)
    def add_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([16, 32, 64]))
        if self.hidden_state_size + numb_new_nodes <= self.max_hidden_state_size:
            self.hidden_state_size += numb_new_nodes
            self.recreate_network()
        return {"numb_new_nodes": numb_new_nodes}

    @mutation(MutationType.NODE)
    def remove_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        if numb_new_nodes is None:
            numb_new_nodes = int(np.random.choice([16, 32, 64]))
        if self.hidden_state_size - numb_new_nodes >= self.min_hidden_state_size:
            self.hidden_state_size -= numb_new_nodes
            self.recreate_network()
        return {"numb_new_nodes": numb_new_nodes}
