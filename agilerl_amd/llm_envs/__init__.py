from .base import LLMEnvBase, make_grpo_experiences
from .reasoning import ReasoningGym, TokenReasoningGym

# reference name for the dataset-backed gym (agilerl/llm_envs/base.py:93)
HuggingFaceGym = ReasoningGym
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
