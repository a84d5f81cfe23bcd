from .base import LLMEnvBase, make_grpo_experiences
from .reasoning import ReasoningGym, TokenReasoningGym
from .sft import SFTGym, SyntheticSFTGym
from .preference import PreferenceGym, SyntheticPreferenceGym
from .multiturn import MultiTurnTokenEnv, TokenGuessEnv, SyncMultiTurnVecEnv

__all__ = [
    "LLMEnvBase",
    "make_grpo_experiences",
    "ReasoningGym",
    "TokenReasoningGym",
    "SFTGym",
    "SyntheticSFTGym",
    "PreferenceGym",
    "SyntheticPreferenceGym",
    "MultiTurnTokenEnv",
    "TokenGuessEnv",
    "SyncMultiTurnVecEnv",
]
