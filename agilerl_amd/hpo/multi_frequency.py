"""MF-PBT: multi-frequency population-based training.

Reference parity: ``agilerl/hpo/multi_frequency.py`` (MF-PBT, Doulazmi et
al.): the population is partitioned into sub-populations that evolve at
different frequencies; within a sub-population slots fall into
winner / survivor / open / loser brackets, and migration moves
high-fitness agents asymmetrically from fast to slow sub-populations so
long-horizon learners are protected.  The distributed runtime uses
:meth:`compute_plan` (rank-0 plan + RCCL broadcast, like tournament).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

__all__ = ["MultiFrequencySelection"]


class MultiFrequencySelection:
    def __init__(
        self,
        frequencies: Sequence[int] = (1, 2, 4),
        elitism: bool = True,
        migration_fraction: float = 0.25,
        rng: Optional[np.random.Generator] = None,
        evolution_frequency_ratios: Optional[Sequence[int]] = None,
        population_size: Optional[int] = None,
        n_subpopulations: Optional[int] = None,
        n_winners: Optional[int] = None,
        n_losers: Optional[int] = None,
        n_survivors: Optional[int] = None,
        n_open_for_migration: Optional[int] = None,
        seed: Optional[int] = None,
    ):
        # reference hpo/multi_frequency.py spellings: frequency ratios are
        # the same thing as `frequencies`; the n_* counts fold into the
        # migration fraction (winners migrate into open slots)
        if evolution_frequency_ratios is not None:
            frequencies = evolution_frequency_ratios
        if n_subpopulations is not None and len(tuple(frequencies)) != int(n_subpopulations):
            raise ValueError(
                f"n_subpopulations={n_subpopulations} does not match "
                f"{len(tuple(frequencies))} frequencies"
            )
        self.population_size = population_size
        if n_winners is not None and n_open_for_migration:
            migration_fraction = min(1.0, n_winners / max(n_open_for_migration, 1))
        self.n_winners = n_winners
        self.n_losers = n_losers
        self.n_survivors = n_survivors
        self.n_open_for_migration = n_open_for_migration
        if seed is not None and rng is None:
            rng = np.random.default_rng(seed)
        self.frequencies = tuple(int(f) for f in frequencies)
        self.elitism = elitism
        self.migration_fraction = migration_fraction
        self.rng = rng or np.random.default_rng(np.random.randint(0, 2**31 - 1))
        self.generation = 0

    # ------------------------------------------------------------------
    def subpop_of(self, slot: int, pop_size: int) -> int:
        """Round-robin frequency assignment over slots."""
        n_sub = len(self.frequencies)
        per = max(pop_size // n_sub, 1)
        return min(slot // per, n_sub - 1)

    def compute_plan(self, fitnesses: np.ndarray, pop_size: Optional[int] = None) -> List[int]:
        """Per-slot parent index for this generation (slots in inactive
        sub-populations keep themselves)."""
        pop_size = pop_size or len(fitnesses)
        self.generation += 1
        plan = list(range(pop_size))

        subpops: Dict[int, List[int]] = {}
        for slot in range(pop_size):
            subpops.setdefault(self.subpop_of(slot, pop_size), []).append(slot)

        global_best = int(np.argmax(fitnesses))
        for sub_idx, slots in subpops.items():
            freq = self.frequencies[sub_idx]
            if self.generation % freq != 0:
                continue  # this sub-population does not evolve this round
            order = sorted(slots, key=lambda s: fitnesses[s], reverse=True)
            n = len(order)
            n_win = max(1, int(np.ceil(n * 0.25)))
            n_lose = max(1, int(np.floor(n * 0.25))) if n > 2 else 0
            winners = order[:n_win]
            losers = order[n - n_lose :] if n_lose else []
            # losers are replaced by clones of (uniformly sampled) winners
            for slot in losers:
                plan[slot] = int(self.rng.choice(winners))
            # asymmetric migration: the globally best agent seeds one open
            # slot of slower sub-populations when it beats their best
            if self.migration_fraction > 0 and global_best not in slots:
                sub_best = order[0]
                if fitnesses[global_best] > fitnesses[sub_best] and n > 1:
                    migrate_slot = order[min(n_win, n - 1)]
                    plan[migrate_slot] = global_best
        return plan

    # ------------------------------------------------------------------
    def select(self, population) -> Tuple[object, List[object]]:
        fitnesses = np.array(
            [a.fitness[-1] if a.fitness else -np.inf for a in population], dtype=np.float64
        )
        plan = self.compute_plan(fitnesses, len(population))
        elite = population[int(np.argmax(fitnesses))]
        new_population = []
        for new_idx, parent_idx in enumerate(plan):
            if parent_idx == new_idx:
                population[new_idx].index = new_idx
                new_population.append(population[new_idx])
            else:
                new_population.append(population[parent_idx].clone(index=new_idx))
        return elite, new_population

    def active_subpops(self) -> List[int]:
        return [i for i, f in enumerate(self.frequencies) if (self.generation + 1) % f == 0]
