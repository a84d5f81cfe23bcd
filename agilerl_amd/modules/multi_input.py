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
        sub_configs: Optional[Dict[str, dict]] = None,
        init_dicts: Optional[Dict[str, dict]] = None,
        vector_space_mlp: bool = True,
        output_activation: Optional[str] = None,
        output_layernorm: bool = False,
        min_latent_dim: int = 8,
        max_latent_dim: int = 128,
        device: str = "cpu",
        name: Optional[str] = None,
        random_seed: Optional[int] = None,
    ):
        super().__init__(device, name=name, random_seed=random_seed)
        # reference multi_input.py:122: vector_space_mlp=False passes vector
        # sub-spaces through a raw flatten (no learned encoder); our default
        # keeps the MLP (documented divergence - richer and mutable).
        # init_dicts are the reference's per-key construction overrides and
        # merge into sub_configs; min/max_latent_dim bound latent mutations.
        self.vector_space_mlp = bool(vector_space_mlp)
        self.output_activation = output_activation
        self.output_layernorm = bool(output_layernorm)
        self.min_latent_dim = int(min_latent_dim)
        self.max_latent_dim = int(max_latent_dim)
        if init_dicts:
            merged = {k: dict(v) for k, v in (sub_configs or {}).items()}
            for k, v in init_dicts.items():
                merged.setdefault(k, {}).update(v)
            sub_configs = merged
        self.observation_space = observation_space
        self.num_outputs = int(num_outputs)
        self.latent_dim = int(latent_dim)
        self.mlp_config = dict(mlp_config or {})
        self.cnn_config = dict(cnn_config or {})
        # per-key overrides: mutations let sub-encoders diverge (each
        # samples its own layer/channel choice), so clones/checkpoints
        # rebuild from the LIVE per-encoder shapes via the `sub_configs`
        # property, not the shared construction-time configs
        ctor_sub = {k: dict(v) for k, v in (sub_configs or {}).items()}

        if isinstance(observation_space, DictSpace):
            items = list(observation_space.spaces.items())
        else:
            items = [(str(i), s) for i, s in enumerate(observation_space.spaces)]
        self._keys = [k for k, _ in items]
        self._subspaces: Dict[str, Space] = dict(items)

        encoders = {}
        for key, space in items:
            encoders[key] = self._make_encoder(space, ctor_sub.get(key))
        self.encoders = nn.ModuleDict(encoders)
        self.head = self._make_head(len(items))

    def _make_head(self, n_items: int) -> nn.Module:
        from .components import get_activation

        layers = [nn.Linear(self.latent_dim * n_items, self.num_outputs)]
        if self.output_layernorm:
            layers.append(nn.LayerNorm(self.num_outputs))
        if self.output_activation:
            layers.append(get_activation(self.output_activation))
        return (layers[0] if len(layers) == 1 else nn.Sequential(*layers)).to(self.device)

    @property
    def sub_configs(self) -> Dict[str, dict]:
        """Live per-encoder architecture (feeds init_dict -> clone)."""
        out: Dict[str, dict] = {}
        for key, enc in self.encoders.items():
            if isinstance(enc, EvolvableCNN):
                out[key] = {
                    "channel_size": list(enc.channel_size),
                    "kernel_size": list(enc.kernel_size),
                    "stride_size": list(enc.stride_size),
                }
            elif isinstance(enc, EvolvableMLP):
                out[key] = {"hidden_size": list(enc.hidden_size)}
            else:  # _FlattenEncoder has no mutable shape
                out[key] = {}
        return out

    def _make_encoder(self, space: Space, override: Optional[dict] = None) -> EvolvableModule:
        if is_image_space(space):
            cfg = {"channel_size": [32, 32], "kernel_size": [3, 3], "stride_size": [2, 2]}
            cfg.update(self.cnn_config)
            cfg.update(override or {})
            return EvolvableCNN(
                input_shape=space.shape, num_outputs=self.latent_dim, device=self.device, **cfg
            )
        if not self.vector_space_mlp:
            return _FlattenEncoder(flatdim(space), self.latent_dim, self.device)
        cfg = {"hidden_size": [64]}
        cfg.update(self.mlp_config)
        cfg.update(override or {})
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
        new_head = self._make_head(len(self._keys))
        preserve_parameters(self.head, new_head)
        self.head = new_head

    @mutation(MutationType.NODE)
    def add_node(
        self, per_encoder: Optional[dict] = None, numb_new_nodes: Optional[int] = None
    ) -> dict:
        return self._fanout("add_node", "add_channel", per_encoder, numb_new_nodes)

    @mutation(MutationType.NODE)
    def remove_node(
        self, per_encoder: Optional[dict] = None, numb_new_nodes: Optional[int] = None
    ) -> dict:
        return self._fanout("remove_node", "remove_channel", per_encoder, numb_new_nodes)

    def _fanout(self, mlp_method, cnn_method, per_encoder, numb_new_nodes) -> dict:
        """Apply a node mutation to every sub-encoder, capturing EVERY
        sampled choice per encoder key so group replay (targets/critics)
        reproduces the exact same shapes — forwarding only the node count
        let each replayed sub-encoder re-sample its layer/channel choice
        and desynchronize from the policy (caught by the arch sweep)."""
        per_encoder = {k: dict(v) for k, v in (per_encoder or {}).items()}
        for key, enc in self.encoders.items():
            method = mlp_method if hasattr(enc, mlp_method) else cnn_method
            choices = per_encoder.get(key)
            if choices is None:
                choices = {}
                if numb_new_nodes is not None:
                    count_key = (
                        "numb_new_nodes" if method.endswith("node") else "numb_new_channels"
                    )
                    choices[count_key] = numb_new_nodes
            r = enc.apply_mutation(method, **choices) or {}
            per_encoder[key] = r
        return {"per_encoder": per_encoder}

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


class _FlattenEncoder(EvolvableModule):
    """vector_space_mlp=False sub-encoder: flatten + single linear projection
    to the shared latent width (no hidden layers, no mutations) — the
    MI355X analog of the reference's raw vector concat
    (multi_input.py:91-94)."""

    def __init__(self, num_inputs: int, num_outputs: int, device: str = "cpu"):
        super().__init__(device)
        self.num_inputs = int(num_inputs)
        self.num_outputs = int(num_outputs)
        self.proj = nn.Linear(self.num_inputs, self.num_outputs).to(device)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not isinstance(x, torch.Tensor):
            x = torch.as_tensor(x, device=self.device)
        x = x.float().reshape(x.shape[0], -1) if x.dim() > 1 else x.float().unsqueeze(0)
        return self.proj(x)

    @property
    def output_size(self) -> int:
        return self.num_outputs

    def reset_noise(self) -> None:
        pass

    def recreate_network(self) -> None:
        pass
