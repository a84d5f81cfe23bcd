"""NDJSON event streaming for Arena experiments.

Reference parity: ``agilerl-arena/agilerl/arena/stream.py``
(``NDJsonStream`` / ``StreamEvent``): experiment progress travels as
newline-delimited JSON events; the client parses them into typed events
and dispatches to a registered handler.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Dict, Iterable, Iterator, Optional

__all__ = ["StreamEvent", "NDJsonStream"]


@dataclass
class StreamEvent:
    """One experiment progress event."""

    kind: str                      # "status" | "metrics" | "log" | "done"
    experiment_id: str = ""
    payload: Dict[str, Any] = field(default_factory=dict)
    ts: Optional[float] = None

    def to_json(self) -> str:
        return json.dumps({
            "kind": self.kind, "experiment_id": self.experiment_id,
            "payload": self.payload, "ts": self.ts,
        })

    @classmethod
    def from_json(cls, line: str) -> "StreamEvent":
        d = json.loads(line)
        return cls(
            kind=d.get("kind", "log"),
            experiment_id=d.get("experiment_id", ""),
            payload=d.get("payload", {}) or {},
            ts=d.get("ts"),
        )


class NDJsonStream:
    """Iterate :class:`StreamEvent`s out of an NDJSON line source
    (an httpx streaming response, a file object, or any line iterable).
    Blank lines and malformed records are skipped (transport keepalives)."""

    def __init__(self, lines: Iterable[str]):
        self._lines = lines

    def __iter__(self) -> Iterator[StreamEvent]:
        for line in self._lines:
            if isinstance(line, bytes):
                line = line.decode("utf-8", errors="replace")
            line = line.strip()
            if not line:
                continue
            try:
                yield StreamEvent.from_json(line)
            except (json.JSONDecodeError, TypeError):
                continue
