from .tournament import TournamentSelection
from .mutation import Mutations

__all__ = ["TournamentSelection", "Mutations"]
