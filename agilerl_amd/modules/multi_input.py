"""Evolvable encoder for Dict / Tuple observation spaces.

Reference parity: ``agilerl/modules/multi_input.py:72`` (EvolvableMultiInput).
One sub-encoder per sub-space (CNN for images, MLP for vectors), features
concatenated then projected to ``num_outputs``.  Mutations broadcast to all
sub-encoders so offspring stay structurally consistent.
"""

from __future__ import annotations

from typing import Dict, Optional, Union

import torch
import torch.nn as nn

from ..spaces import DictSpace, Space, TupleSpace, flatdim, is_image_space
from .base import EvolvableModule, MutationType, mutation, preserve_parameters
from .cnn import EvolvableCNN
from .mlp import EvolvableMLP

__all__ = ["EvolvableMultiInput"]


class EvolvableMultiInput(EvolvableModule):
    def __init__(
        self,
        observation_space: Union[DictSpace, TupleSpace],
        num_outputs: int,
        latent_dim: int = 64,
        mlp_config: Optional[dict] = None,
        cnn_config: Optional[dict] = None,
        device: str = "cpu",
    ):
        super().__init__(device)
        self.observation_space = observation_space
        self.num_outputs = int(num_outputs)
        self.latent_dim = int(latent_dim)
        self.mlp_config = dict(mlp_config or {})
        self.cnn_config = dict(cnn_config or {})

        if isinstance(observation_space, DictSpace):
            items = list(observation_space.spaces.items())
        else:
            items = [(str(i), s) for i, s in enumerate(observation_space.spaces)]
        self._keys = [k for k, _ in items]
        self._subspaces: Dict[str, Space] = dict(items)

        encoders = {}
        for key, space in items:
            encoders[key] = self._make_encoder(space)
        self.encoders = nn.ModuleDict(encoders)
        self.head = nn.Linear(self.latent_dim * len(items), self.num_outputs).to(device)

    def _make_encoder(self, space: Space) -> EvolvableModule:
        if is_image_space(space):
            cfg = {"channel_size": [32, 32], "kernel_size": [3, 3], "stride_size": [2, 2]}
            cfg.update(self.cnn_config)
            return EvolvableCNN(
                input_shape=space.shape, num_outputs=self.latent_dim, device=self.device, **cfg
            )
        cfg = {"hidden_size": [64]}
        cfg.update(self.mlp_config)
        return EvolvableMLP(
            num_inputs=flatdim(space), num_outputs=self.latent_dim, device=self.device, **cfg
        )

    def forward(self, obs) -> torch.Tensor:
        feats = []
        for key in self._keys:
            x = obs[key] if isinstance(obs, dict) else obs[int(key)]
            feats.append(self.encoders[key](x))
        return self.head(torch.cat(feats, dim=-1))

    @property
    def output_size(self) -> int:
        return self.num_outputs

    def reset_noise(self) -> None:
        for enc in self.encoders.values():
            enc.reset_noise()

    # ------------------------------------------------------------------
    def recreate_network(self) -> None:
        for enc in self.encoders.values():
            enc.recreate_network()
        new_head = nn.Linear(self.latent_dim * len(self._keys), self.num_outputs).to(self.device)
        preserve_parameters(self.head, new_head)
        self.head = new_head

    @mutation(MutationType.NODE)
    def add_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        out = {}
        for enc in self.encoders.values():
            method = "add_node" if hasattr(enc, "add_node") else "add_channel"
            r = enc.apply_mutation(method, **out) or {}
            nn_key = "numb_new_nodes" if "numb_new_nodes" in r else None
            if nn_key and "numb_new_nodes" not in out:
                out["numb_new_nodes"] = r["numb_new_nodes"]
        return {"numb_new_nodes": out.get("numb_new_nodes", numb_new_nodes)}

    @mutation(MutationType.NODE)
    def remove_node(self, numb_new_nodes: Optional[int] = None) -> dict:
        out = {}
        for enc in self.encoders.values():
            method = "remove_node" if hasattr(enc, "remove_node") else "remove_channel"
            r = enc.apply_mutation(method, **out) or {}
            if "numb_new_nodes" in r and "numb_new_nodes" not in out:
                out["numb_new_nodes"] = r["numb_new_nodes"]
        return {"numb_new_nodes": out.get("numb_new_nodes", numb_new_nodes)}

    @mutation(MutationType.ACTIVATION)
    def change_activation(self, activation: Optional[str] = None, output: bool = False) -> dict:
        r: dict = {}
        for enc in self.encoders.values():
            if hasattr(enc, "change_activation"):
                r = enc.apply_mutation(
                    "change_activation", activation=activation or r.get("activation"), output=output
                ) or r
                activation = r.get("activation", activation)
        return {"activation": activation, "output": output}
