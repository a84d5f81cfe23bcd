"""Language-environment ABCs for the offline token-RL (ILQL/BC) stack.

Reference parity: ``agilerl/data/language_environment.py:12-72``
(Language_Observation / Language_Environment / Policy /
interact_environment).
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Dict, List, Optional, Tuple

__all__ = ["LanguageObservation", "LanguageEnvironment", "LanguagePolicy",
           "interact_environment"]


class LanguageObservation(ABC):
    """A dialogue state: alternating utterances, agent turns carry rewards."""

    @abstractmethod
    def to_sequence(self) -> Tuple[List[Tuple[str, Optional[float]]], bool]:
        """Returns ([(utterance, reward-or-None), ...], terminal).  A reward
        of None marks an environment utterance; a float marks an agent
        action and its reward."""

    def metadata(self) -> Optional[Dict[str, Any]]:
        return None


class LanguageEnvironment(ABC):
    @abstractmethod
    def reset(self) -> LanguageObservation: ...

    @abstractmethod
    def step(self, action: str) -> Tuple[LanguageObservation, float, bool]: ...

    def is_terminal(self, obs: LanguageObservation) -> bool:
        return obs.to_sequence()[1]


class LanguagePolicy(ABC):
    @abstractmethod
    def act(self, obs: LanguageObservation) -> str: ...

    def train(self) -> None:  # optional mode switches
        pass

    def eval(self) -> None:
        pass


def interact_environment(env: LanguageEnvironment, policy: LanguagePolicy,
                         obs: Optional[LanguageObservation] = None,
                         max_turns: int = 32):
    """Roll one episode: policy acts until the env reports terminal.
    Returns (final_obs, total_reward, turns)."""
    if obs is None:
        obs = env.reset()
    total, turns = 0.0, 0
    while not env.is_terminal(obs) and turns < max_turns:
        action = policy.act(obs)
        obs, reward, done = env.step(action)
        total += float(reward)
        turns += 1
        if done:
            break
    return obs, total, turns
