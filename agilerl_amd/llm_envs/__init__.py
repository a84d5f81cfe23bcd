from .base import LLMEnvBase, make_grpo_experiences
from .reasoning import ReasoningGym, TokenReasoningGym
from .sft import SFTGym, SyntheticSFTGym
from .preference import PreferenceGym, SyntheticPreferenceGym

__all__ = [
    "LLMEnvBase",
    "make_grpo_experiences",
    "ReasoningGym",
    "TokenReasoningGym",
    "SFTGym",
    "SyntheticSFTGym",
    "PreferenceGym",
    "SyntheticPreferenceGym",
]
