"""Tool use + format-reward utilities for LLM envs.

Reference parity: ``agilerl/llm_envs/search.py`` (SearchTool :21,
FormatRewardWrapper :111).  The search tool serves offline corpora (no
network); the format wrapper shapes rewards toward a required output
structure (e.g. ``<answer>...</answer>`` tags).
"""

from __future__ import annotations

import re
from typing import Callable, List, Optional, Sequence

import numpy as np

__all__ = ["SearchTool", "FormatRewardWrapper", "extract_answer"]


ANSWER_RE = re.compile(r"<answer>(.*?)</answer>", re.DOTALL)


def extract_answer(text: str) -> Optional[str]:
    m = ANSWER_RE.search(text)
    return m.group(1).strip() if m else None


class SearchTool:
    """Keyword search over an offline document list.

    ``__call__(query, k)`` returns the top-k documents by term overlap —
    the retrieval primitive a multi-turn agent can invoke between turns.
    """

    def __init__(self, documents: Sequence[str]):
        self.documents = list(documents)
        self._tokens = [set(d.lower().split()) for d in self.documents]

    def __call__(self, query: str, k: int = 3) -> List[str]:
        q = set(query.lower().split())
        scores = [len(q & toks) for toks in self._tokens]
        order = np.argsort(scores)[::-1][:k]
        return [self.documents[i] for i in order if scores[i] > 0]


class FormatRewardWrapper:
    """Wraps a text reward_fn: adds a format bonus when the completion
    carries a well-formed ``<answer>`` block and evaluates the inner
    answer (instead of the raw completion) with the base reward."""

    def __init__(
        self,
        reward_fn: Callable[[str, object], float],
        format_bonus: float = 0.1,
        require_format: bool = False,
    ):
        self.reward_fn = reward_fn
        self.format_bonus = format_bonus
        self.require_format = require_format

    def __call__(self, completion: str, answer) -> float:
        extracted = extract_answer(completion)
        if extracted is None:
            if self.require_format:
                return 0.0
            return float(self.reward_fn(completion, answer))
        return float(self.reward_fn(extracted, answer)) + self.format_bonus


TOOL_CALL_RE = re.compile(r"<tool>(.*?)</tool>", re.DOTALL)


def parse_tool_calls(text: str) -> List[str]:
    """Extract ``<tool>query</tool>`` invocations from a completion
    (reference search-env tool-call convention)."""
    return [m.strip() for m in TOOL_CALL_RE.findall(text)]
