"""Shared type aliases.

Reference parity: ``agilerl/typing.py`` (503 LoC of aliases) — the subset
this framework's public surface uses.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Tuple, Union

import numpy as np
import torch

from .spaces import Space

# observations -------------------------------------------------------------
ArrayLike = Union[np.ndarray, torch.Tensor]
ObservationType = Union[ArrayLike, Dict[str, ArrayLike], Tuple[ArrayLike, ...]]
MultiAgentObservationType = Dict[str, ObservationType]

# actions ------------------------------------------------------------------
ActionType = Union[int, np.ndarray, torch.Tensor]
MultiAgentActionType = Dict[str, ActionType]

# experiences --------------------------------------------------------------
ExperiencesType = Dict[str, Any]
MultiAgentExperiencesType = Dict[str, Dict[str, Any]]

# config -------------------------------------------------------------------
NetConfigType = Dict[str, Any]
KwargsType = Dict[str, Any]
DeviceType = Union[str, torch.device]

# spaces -------------------------------------------------------------------
SpaceType = Space
MultiAgentSpaceType = Dict[str, Space]

# population ---------------------------------------------------------------
PopulationType = List[Any]
FitnessType = List[float]

# LLM ----------------------------------------------------------------------
TokenBatch = Dict[str, torch.Tensor]
ReasoningPrompts = Union[List[str], TokenBatch]
RewardFnType = Callable[..., float]

# checkpoints --------------------------------------------------------------
CheckpointInfo = Dict[str, Any]
