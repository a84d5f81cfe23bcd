from .tournament import TournamentSelection
from .mutation import Mutations
from .multi_frequency import MultiFrequencySelection

__all__ = ["TournamentSelection", "Mutations", "MultiFrequencySelection"]
