"""Supervised fine-tuning loop.

Reference parity: ``agilerl/training/llm/sft.py:30``.
"""

from __future__ import annotations

import time as _time
from typing import List, Optional

import numpy as np

from ...hpo.mutation import Mutations
from ...hpo.tournament import TournamentSelection
from ...population import Population

__all__ = ["finetune_llm_sft"]


def finetune_llm_sft(
    env,
    pop: List,
    max_steps: int = 1000,
    evo_steps: int = 100,
    eval_loop: int = 1,
    target: Optional[float] = None,
    tournament: Optional[TournamentSelection] = None,
    mutation: Optional[Mutations] = None,
    loggers: Optional[List] = None,
    max_wall_seconds: Optional[float] = None,
    verbose: bool = True,
):
    t_start = _time.time()
    if loggers is None and verbose:
        from ...logger import StdOutLogger

        loggers = [StdOutLogger()]
    population = Population(pop, loggers or [])
    fitness_history: List[List[float]] = []

    while population.all_below(max_steps) and not population.should_stop(target):
        for agent, metrics in zip(population.agents, population.metrics):
            metrics.init_training_step()
            losses = []
            for _ in range(evo_steps):
                batch = env.sample()
                losses.append(agent.learn(batch)["loss"])
                agent.steps[-1] += 1
            metrics.finalize_training_step(evo_steps)
            metrics.log("loss", float(np.mean(losses)))
            fitness = agent.test(env, loop=eval_loop)
            metrics.log_fitness(fitness)
        fitness_history.append([a.fitness[-1] for a in population.agents])
        if verbose or population.loggers:
            population.report_metrics()
        if population.should_stop(target) or not population.all_below(max_steps):
            break
        if tournament is not None and mutation is not None:
            old = list(population.agents)
            elite, new_pop = tournament.select(population.agents)
            new_pop = mutation.mutation(new_pop)
            for agent in old:
                if agent not in new_pop:
                    agent.clean_up()
            population.replace(new_pop)
        population.increment_evo_step()
        if max_wall_seconds is not None and _time.time() - t_start > max_wall_seconds:
            break
    return population.agents, fitness_history
