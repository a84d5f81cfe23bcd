"""Dialogue tokenizer contract for the offline token-RL stack.

Reference parity: ``agilerl/data/tokenizer.py:8`` (Tokenizer ABC with
dialogue boundary tokens).  Any tokenizer exposing ``encode``/``decode``
plus the five boundary ids works; :class:`DialogueTokenizer` adapts a
plain vocabulary (or an HF tokenizer) by appending the boundary tokens
to the id space.
"""

from __future__ import annotations

from typing import List, Optional

__all__ = ["DialogueTokenizer"]


class DialogueTokenizer:
    """Wraps a base tokenizer (or builds a byte-level one) and reserves
    five dialogue-boundary ids: bos (dialogue starts with env turn),
    boa (dialogue starts with agent turn), eos (end of env utterance),
    eoa (end of agent action), eod (terminal)."""

    SPECIALS = ("<bos>", "<boa>", "<eos>", "<eoa>", "<eod>")

    def __init__(self, base=None, vocab_size: Optional[int] = None):
        self.base = base
        if base is not None:
            self._base_size = int(getattr(base, "vocab_size", len(base)))
        else:
            self._base_size = int(vocab_size or 256)  # byte-level fallback
        ids = range(self._base_size, self._base_size + 5)
        (self.bos_token_id, self.boa_token_id, self.eos_token_id,
         self.eoa_token_id, self.eod_token_id) = ids
        self.vocab_size = self._base_size + 5
        self._special_by_id = dict(zip(ids, self.SPECIALS))
        self._special_by_tok = {v: k for k, v in self._special_by_id.items()}

    # -- single-string encode/decode with boundary-token passthrough ----
    def encode(self, text: str) -> List[int]:
        out: List[int] = []
        i = 0
        while i < len(text):
            matched = False
            for tok, tid in self._special_by_tok.items():
                if text.startswith(tok, i):
                    out.append(tid)
                    i += len(tok)
                    matched = True
                    break
            if matched:
                continue
            if self.base is not None:
                # longest non-special prefix through the base tokenizer
                j = i
                while j < len(text) and not any(
                    text.startswith(t, j) for t in self._special_by_tok
                ):
                    j += 1
                out.extend(self.base.encode(text[i:j], add_special_tokens=False)
                           if hasattr(self.base, "encode") else self.base(text[i:j]))
                i = j
            else:
                out.append(min(ord(text[i]), self._base_size - 1))
                i += 1
        return out

    def decode(self, ids: List[int]) -> str:
        parts: List[str] = []
        plain: List[int] = []

        def flush():
            if not plain:
                return
            if self.base is not None:
                parts.append(self.base.decode(plain))
            else:
                parts.append("".join(chr(t) for t in plain))
            plain.clear()

        for t in ids:
            if t in self._special_by_id:
                flush()
                parts.append(self._special_by_id[t])
            else:
                plain.append(int(t))
        flush()
        return "".join(parts)

    def id_to_token(self, tid: int) -> str:
        return self._special_by_id.get(tid, self.decode([tid]))
