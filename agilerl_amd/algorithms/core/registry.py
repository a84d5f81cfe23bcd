"""Mutation registry: network groups, optimizer configs, hyperparameter specs.

Reference parity: ``agilerl/algorithms/core/registry.py`` — NetworkGroup
:273, OptimizerConfig :55, RLParameter :134, HyperparameterConfig :190,
MutationRegistry :440.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import numpy as np

__all__ = [
    "NetworkGroup",
    "OptimizerConfig",
    "RLParameter",
    "HyperparameterConfig",
    "MutationRegistry",
]


@dataclass
class NetworkGroup:
    """A group of structurally-tied networks.

    ``eval_network`` is the attribute name of the trained network;
    ``shared_networks`` are attribute names of networks that must mirror its
    architecture (target networks, frozen copies).  Exactly one group per
    agent has ``policy=True`` — architecture mutations are sampled from its
    eval network's mutation surface.
    """

    eval_network: str
    shared_networks: List[str] = field(default_factory=list)
    policy: bool = False
    multiagent: bool = False

    def all_names(self) -> List[str]:
        return [self.eval_network] + list(self.shared_networks)


@dataclass
class OptimizerConfig:
    """Description of one optimizer the agent owns.

    ``networks``: attribute names whose parameters this optimizer trains.
    ``lr_name``: agent attribute holding the (mutable) learning rate.
    """

    name: str
    networks: List[str]
    lr_name: str = "lr"
    optimizer_cls: Any = None  # default torch.optim.Adam, resolved lazily
    optimizer_kwargs: Dict[str, Any] = field(default_factory=dict)


class RLParameter:
    """A mutable RL hyperparameter with bounds and mutation semantics."""

    def __init__(
        self,
        min: float,
        max: float,
        dtype: type = float,
        shrink_factor: float = 0.8,
        grow_factor: float = 1.2,
        categorical: Optional[List[Any]] = None,
    ):
        self.min = min
        self.max = max
        self.dtype = dtype
        self.shrink_factor = shrink_factor
        self.grow_factor = grow_factor
        self.categorical = categorical

    def mutate(self, value, rng: Optional[np.random.Generator] = None):
        rng = rng or np.random.default_rng(np.random.randint(0, 2**31 - 1))
        if self.categorical is not None:
            return self.categorical[int(rng.integers(len(self.categorical)))]
        factor = self.grow_factor if rng.random() < 0.5 else self.shrink_factor
        new = np.clip(value * factor, self.min, self.max)
        if self.dtype is int:
            new = int(round(float(new)))
            if new == int(value):
                new = int(np.clip(new + (1 if factor > 1 else -1), self.min, self.max))
            return new
        return float(new)


class HyperparameterConfig:
    """name -> RLParameter mapping; names are agent attribute names."""

    def __init__(self, **params: RLParameter):
        self.config: Dict[str, RLParameter] = dict(params)

    def names(self) -> List[str]:
        return list(self.config.keys())

    def sample(self, rng: Optional[np.random.Generator] = None) -> Optional[str]:
        if not self.config:
            return None
        rng = rng or np.random.default_rng(np.random.randint(0, 2**31 - 1))
        return self.names()[int(rng.integers(len(self.config)))]

    def __getitem__(self, name: str) -> RLParameter:
        return self.config[name]

    def __contains__(self, name: str) -> bool:
        return name in self.config

    def __bool__(self) -> bool:
        return bool(self.config)

    def items(self):
        return self.config.items()


class MutationRegistry:
    """Per-agent registry of groups, optimizers, and mutation hooks."""

    def __init__(self):
        self.groups: List[NetworkGroup] = []
        self.optimizer_configs: List[OptimizerConfig] = []
        self.hooks: List[str] = []  # method names on the agent

    def register_group(self, group: NetworkGroup) -> None:
        self.groups.append(group)

    def register_optimizer(self, config: OptimizerConfig) -> None:
        self.optimizer_configs.append(config)

    def register_hook(self, method_name: str) -> None:
        if method_name not in self.hooks:
            self.hooks.append(method_name)

    @property
    def policy_group(self) -> Optional[NetworkGroup]:
        for g in self.groups:
            if g.policy:
                return g
        return self.groups[0] if self.groups else None

    def all_network_names(self) -> List[str]:
        names: List[str] = []
        for g in self.groups:
            for n in g.all_names():
                if n not in names:
                    names.append(n)
        return names

    def eval_network_names(self) -> List[str]:
        names: List[str] = []
        for g in self.groups:
            if g.eval_network not in names:
                names.append(g.eval_network)
        return names
