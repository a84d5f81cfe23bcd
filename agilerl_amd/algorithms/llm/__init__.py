from .base import LLMAlgorithm, build_causal_lm
from .grpo import GRPO
from .gspo import GSPO
from .cispo import CISPO
from .sft import SFT
from .dpo import DPO
from .ppo_llm import PPOLLM
from .reinforce_llm import ReinforceLLM

__all__ = ["LLMAlgorithm", "build_causal_lm", "GRPO", "GSPO", "CISPO", "SFT", "DPO", "PPOLLM", "ReinforceLLM"]
