"""Structural interfaces (Protocol classes).

Reference parity: ``agilerl/protocols.py`` (23 Protocol classes breaking
import cycles).  These document the duck-typed contracts between layers;
isinstance checks use ``runtime_checkable``.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Protocol, Tuple, runtime_checkable

import numpy as np
import torch

__all__ = [
    "EvolvableModuleProtocol",
    "EvolvableNetworkProtocol",
    "EvolvableAlgorithmProtocol",
    "MultiAgentAlgorithmProtocol",
    "LLMAlgorithmProtocol",
    "AgentWrapperProtocol",
    "VecEnvProtocol",
    "MultiAgentVecEnvProtocol",
    "ReplayBufferProtocol",
    "RolloutBufferProtocol",
    "SelectionProtocol",
    "MutationsProtocol",
    "LoggerProtocol",
    "MetricsProtocol",
    "LLMEnvProtocol",
]


@runtime_checkable
class EvolvableModuleProtocol(Protocol):
    device: str

    @property
    def init_dict(self) -> Dict[str, Any]: ...

    @property
    def mutation_methods(self) -> List[str]: ...

    def apply_mutation(self, name: str, **choices) -> Optional[dict]: ...

    def clone(self) -> "EvolvableModuleProtocol": ...


@runtime_checkable
class EvolvableNetworkProtocol(EvolvableModuleProtocol, Protocol):
    latent_dim: int

    def forward(self, obs) -> torch.Tensor: ...


@runtime_checkable
class EvolvableAlgorithmProtocol(Protocol):
    index: int
    fitness: List[float]
    steps: List[int]
    mut: str

    def get_action(self, obs, **kwargs): ...

    def learn(self, experiences, **kwargs): ...

    def test(self, env, **kwargs) -> float: ...

    def clone(self, index: Optional[int] = None, wrap: bool = True): ...

    def save_checkpoint(self, path: str) -> None: ...

    def load_checkpoint(self, path: str) -> None: ...


@runtime_checkable
class MultiAgentAlgorithmProtocol(EvolvableAlgorithmProtocol, Protocol):
    agent_ids: List[str]
    n_agents: int


@runtime_checkable
class LLMAlgorithmProtocol(Protocol):
    adapter_name: str

    def generate(self, input_ids, attention_mask, **kwargs) -> torch.Tensor: ...

    def compute_logprobs(self, input_ids, attention_mask, **kwargs) -> torch.Tensor: ...


@runtime_checkable
class AgentWrapperProtocol(Protocol):
    agent: Any

    def get_action(self, obs, *args, **kwargs): ...

    def learn(self, experiences, *args, **kwargs): ...


@runtime_checkable
class VecEnvProtocol(Protocol):
    num_envs: int

    def reset(self, seed: Optional[int] = None): ...

    def step(self, actions): ...


@runtime_checkable
class MultiAgentVecEnvProtocol(Protocol):
    num_envs: int
    agents: List[str]

    def reset(self, seed: Optional[int] = None): ...

    def step(self, actions: Dict[str, Any]): ...


@runtime_checkable
class ReplayBufferProtocol(Protocol):
    def add(self, *args, **kwargs) -> None: ...

    def sample(self, batch_size: int, **kwargs): ...

    def __len__(self) -> int: ...


@runtime_checkable
class RolloutBufferProtocol(Protocol):
    def add(self, **kwargs) -> None: ...

    def compute_returns_and_advantages(self, last_value, last_done=None) -> None: ...

    def get_minibatches(self, batch_size: int, shuffle: bool = True): ...


@runtime_checkable
class SelectionProtocol(Protocol):
    def select(self, population) -> Tuple[Any, List[Any]]: ...


@runtime_checkable
class MutationsProtocol(Protocol):
    def mutation(self, population, pre_training: bool = False) -> List[Any]: ...


@runtime_checkable
class LoggerProtocol(Protocol):
    def log_report(self, report: Dict[str, Any]) -> None: ...


@runtime_checkable
class MetricsProtocol(Protocol):
    def log(self, name: str, value: float) -> None: ...

    def snapshot(self, agent=None) -> Dict[str, Any]: ...


@runtime_checkable
class LLMEnvProtocol(Protocol):
    group_size: int
    prompt_len: int

    def reset(self) -> Dict[str, torch.Tensor]: ...

    def score(self, sequences: torch.Tensor) -> np.ndarray: ...
