"""Tournament selection with elitism.

Reference parity: ``agilerl/hpo/tournament.py:20`` (TournamentSelection,
``select`` :76).  The distributed one-agent-per-GPU path performs the
selection on rank 0 and broadcasts the clone plan (see
``agilerl_amd/parallel/population_runtime.py`` — RCCL broadcast over xGMI
replacing the reference's pickled ``broadcast_object_list``).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np

__all__ = ["TournamentSelection"]


class TournamentSelection:
    def __init__(
        self,
        tournament_size: int = 2,
        elitism: bool = True,
        population_size: Optional[int] = None,
        eval_loop: int = 1,
        rng: Optional[np.random.Generator] = None,
    ):
        self.tournament_size = int(tournament_size)
        self.elitism = bool(elitism)
        self.population_size = population_size
        self.eval_loop = eval_loop
        self.rng = rng or np.random.default_rng(np.random.randint(0, 2**31 - 1))

    # ------------------------------------------------------------------
    def _fitnesses(self, population) -> np.ndarray:
        return np.array(
            [
                np.mean(agent.fitness[-self.eval_loop :]) if agent.fitness else -np.inf
                for agent in population
            ],
            dtype=np.float64,
        )

    def compute_plan(self, fitnesses: np.ndarray, pop_size: Optional[int] = None) -> List[int]:
        """Pure selection logic: returns, per offspring slot, the parent index.

        Slot 0 is the elite when elitism is on.  Separated from agent
        cloning so the distributed runtime can broadcast just this plan.
        """
        pop_size = pop_size or self.population_size or len(fitnesses)
        rank = np.argsort(fitnesses)  # ascending
        elite_idx = int(rank[-1])
        plan: List[int] = []
        if self.elitism:
            plan.append(elite_idx)
        while len(plan) < pop_size:
            k = min(self.tournament_size, len(fitnesses))
            contenders = self.rng.choice(len(fitnesses), size=k, replace=False)
            winner = int(contenders[np.argmax(fitnesses[contenders])])
            plan.append(winner)
        return plan

    def select(self, population) -> Tuple[object, List[object]]:
        """Returns (elite_agent, new_population)."""
        fitnesses = self._fitnesses(population)
        plan = self.compute_plan(fitnesses, len(population))
        elite = population[int(np.argmax(fitnesses))]
        new_population = []
        for new_idx, parent_idx in enumerate(plan):
            clone = population[parent_idx].clone(index=new_idx)
            new_population.append(clone)
        return elite, new_population
