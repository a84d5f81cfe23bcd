"""agilerl-amd: MI355X-native evolutionary-HPO RL training framework.

A from-scratch framework with the capabilities, public API shape and
checkpoint format of AgileRL, built for one 8xMI355X node: PyTorch-ROCm +
hand-written CDNA4 HIP kernels for the hot numeric paths + RCCL
collectives over the xGMI mesh (one population agent per GPU).  No
Accelerate/DeepSpeed/Triton/vLLM dependencies.

Public names mirror the reference (``agilerl/__init__.py:74-79``):
``LocalTrainer`` and the capability flags; algorithm classes live under
``agilerl_amd.algorithms``.
"""

from __future__ import annotations

import importlib
from enum import Enum
from typing import TYPE_CHECKING

__version__ = "0.1.0"

# ---------------------------------------------------------------------------
# Capability flags (reference parity: HAS_* in agilerl/__init__.py:56-60).
# ---------------------------------------------------------------------------


def _importable(name: str) -> bool:
    try:
        importlib.import_module(name)
        return True
    except ImportError:
        return False


HAS_LLM_DEPENDENCIES = _importable("transformers")
HAS_ARENA_DEPENDENCIES = _importable("fastapi") and _importable("httpx")
HAS_HIP_KERNELS = None  # resolved lazily below

# Reference-familiar flags: these stacks are REPLACED by first-party
# MI355X-native code (HIP kernels, RCCL layer, in-framework LoRA), so the
# flags exist for API familiarity and are always False.
HAS_VLLM = False
HAS_DEEPSPEED = False
HAS_LIGER_KERNEL = False


class AgentType(str, Enum):
    RL = "rl"
    MULTI_AGENT_RL = "multi_agent_rl"
    BANDIT = "bandit"
    OFFLINE = "offline"
    LLM = "llm"


_LAZY = {
    "LocalTrainer": "agilerl_amd.training.trainer",
    "Trainer": "agilerl_amd.training.trainer",
    "ArenaTrainer": "agilerl_amd.training.trainer",
    "Population": "agilerl_amd.population",
    "TournamentSelection": "agilerl_amd.hpo.tournament",
    "Mutations": "agilerl_amd.hpo.mutation",
    "MultiFrequencySelection": "agilerl_amd.hpo.multi_frequency",
    "make_vect_envs": "agilerl_amd.envs.registry",
    "create_app": "agilerl_amd.serve",
    "load_agent": "agilerl_amd.serve",
    "create_population": "agilerl_amd.utils.utils",
    "TrainingManifest": "agilerl_amd.models.manifest",
    "ArenaClient": "agilerl_amd.arena.client",
    "DistributedPopulation": "agilerl_amd.parallel.population_runtime",
    "DistributedState": "agilerl_amd.parallel.state",
}


def __getattr__(name: str):
    if name == "HAS_HIP_KERNELS":
        from .ops import has_extension

        return has_extension()
    if name in _LAZY:
        module = importlib.import_module(_LAZY[name])
        return getattr(module, name)
    raise AttributeError(f"module 'agilerl_amd' has no attribute '{name}'")


__all__ = [
    "__version__",
    "AgentType",
    "HAS_LLM_DEPENDENCIES",
    "HAS_ARENA_DEPENDENCIES",
    "HAS_HIP_KERNELS",
    "HAS_VLLM",
    "HAS_DEEPSPEED",
    "HAS_LIGER_KERNEL",
    "LocalTrainer",
    "Trainer",
    "ArenaTrainer",
    "Population",
    "TournamentSelection",
    "Mutations",
    "MultiFrequencySelection",
    "make_vect_envs",
    "create_app",
    "load_agent",
    "create_population",
    "TrainingManifest",
    "ArenaClient",
    "DistributedPopulation",
    "DistributedState",
]
