from .base import HuggingFaceGym, LLMEnvBase, make_grpo_experiences
from .reasoning import ReasoningGym, TokenReasoningGym
from .sft import SFTGym, SyntheticSFTGym
from .preference import PreferenceGym, SyntheticPreferenceGym
from .multiturn import (MultiTurnTokenEnv, TokenGuessEnv, SyncMultiTurnVecEnv,
                        TextMultiTurnEnv, SearchQAEnv)
from .search import SearchTool, FormatRewardWrapper, extract_answer

__all__ = [
    "LLMEnvBase",
    "make_grpo_experiences",
    "ReasoningGym",
    "HuggingFaceGym",
    "TokenReasoningGym",
    "SFTGym",
    "SyntheticSFTGym",
    "PreferenceGym",
    "SyntheticPreferenceGym",
    "MultiTurnTokenEnv",
    "TokenGuessEnv",
    "SyncMultiTurnVecEnv",
    "TextMultiTurnEnv",
    "SearchQAEnv",
    "SearchTool",
    "FormatRewardWrapper",
    "extract_answer",
]
