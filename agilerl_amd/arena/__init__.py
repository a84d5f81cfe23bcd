from .client import ArenaClient, ArenaError, ExperimentHandle
from .stream import NDJsonStream, StreamEvent

__all__ = [
    "ArenaClient",
    "ExperimentHandle",
    "ArenaError",
    "StreamEvent",
    "NDJsonStream",
    "ArenaService",
    "create_app",
]


def __getattr__(name):
    # service pulls in FastAPI; load lazily so the client stays light
    if name in ("ArenaService", "create_app"):
        from . import service

        return getattr(service, name)
    raise AttributeError(name)
