from .client import ArenaClient, ExperimentHandle, ArenaError

__all__ = ["ArenaClient", "ExperimentHandle", "ArenaError"]
