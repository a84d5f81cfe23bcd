from .rl_data import (
    ConstantTokenReward,
    DataPoint,
    IterableRLDataset,
    ListRLDataset,
    SpecifiedTokenReward,
    TokenReward,
)
from .language_environment import (
    LanguageEnvironment,
    LanguageObservation,
    LanguagePolicy,
    interact_environment,
)
from .tokenizer import DialogueTokenizer

__all__ = [
    "TokenReward", "ConstantTokenReward", "SpecifiedTokenReward",
    "DataPoint", "ListRLDataset", "IterableRLDataset",
    "LanguageObservation", "LanguageEnvironment", "LanguagePolicy",
    "interact_environment", "DialogueTokenizer",
]
