"""Chat-template helpers.

Reference parity: ``agilerl/utils/chat_template.py`` and
``llm_envs/base.py:36`` (``apply_chat_template``).  Uses the tokenizer's
own Jinja chat template when present; otherwise a simple fallback format
so offline/random-init models still get consistent prompt structure.
"""

from __future__ import annotations

from typing import Dict, List, Optional

__all__ = ["apply_chat_template", "DEFAULT_TEMPLATE"]

DEFAULT_TEMPLATE = (
    "{sys}\n" "User: {user}\n" "Assistant:"
)


def apply_chat_template(
    tokenizer,
    user_message: str,
    system_prompt: Optional[str] = None,
    add_generation_prompt: bool = True,
) -> str:
    messages: List[Dict[str, str]] = []
    if system_prompt:
        messages.append({"role": "system", "content": system_prompt})
    messages.append({"role": "user", "content": user_message})
    if tokenizer is not None and getattr(tokenizer, "chat_template", None):
        return tokenizer.apply_chat_template(
            messages, tokenize=False, add_generation_prompt=add_generation_prompt
        )
    return DEFAULT_TEMPLATE.format(sys=system_prompt or "", user=user_message)
