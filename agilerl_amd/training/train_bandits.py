"""Contextual-bandit population training loop.

Reference parity: ``agilerl/training/train_bandits.py``.
"""

from __future__ import annotations

import time as _time
from typing import List, Optional

import numpy as np

from ..components.replay_buffer import ReplayBuffer
from ..hpo.mutation import Mutations
from ..hpo.tournament import TournamentSelection
from ..population import Population

__all__ = ["train_bandits"]


def train_bandits(
    env,
    env_name: str,
    algo: str,
    pop: List,
    memory: Optional[ReplayBuffer] = None,
    max_steps: int = 10_000,
    evo_steps: int = 500,
    eval_steps: Optional[int] = 200,
    eval_loop: int = 1,
    target: Optional[float] = None,
    tournament: Optional[TournamentSelection] = None,
    mutation: Optional[Mutations] = None,
    checkpoint: Optional[int] = None,
    checkpoint_path: Optional[str] = None,
    loggers: Optional[List] = None,
    max_wall_seconds: Optional[float] = None,
    verbose: bool = True,
):
    t_start = _time.time()
    if loggers is None and verbose:
        from ..logger import StdOutLogger

        loggers = [StdOutLogger()]
    population = Population(pop, loggers or [])
    memory = memory or ReplayBuffer(10_000)
    fitness_history: List[List[float]] = []

    while population.all_below(max_steps) and not population.should_stop(target):
        for agent, metrics in zip(population.agents, population.metrics):
            metrics.init_training_step()
            context = env.reset()
            total_reward, losses = 0.0, []
            for step in range(evo_steps):
                arm = agent.get_action(context)
                reward, next_context = env.step(arm)
                total_reward += reward
                memory.add(
                    obs=np.asarray(context[arm], dtype=np.float32)[None],
                    reward=np.array([reward], dtype=np.float32),
                )
                context = next_context
                if len(memory) >= agent.batch_size and step % agent.learn_step == 0:
                    losses.append(agent.learn(memory.sample(agent.batch_size)))
            agent.steps[-1] += evo_steps
            metrics.finalize_training_step(evo_steps)
            if losses:
                metrics.log("loss", float(np.mean(losses)))
            metrics.log("mean_regret", float(np.mean(env.regret)) if env.regret else 0.0)
            fitness = agent.test(env, max_steps=eval_steps, loop=eval_loop)
            metrics.log_fitness(fitness)

        fitness_history.append([a.fitness[-1] for a in population.agents])
        if verbose or population.loggers:
            population.report_metrics()
        if population.should_stop(target) or not population.all_below(max_steps):
            break
        if tournament is not None and mutation is not None:
            elite, new_pop = tournament.select(population.agents)
            new_pop = mutation.mutation(new_pop)
            population.replace(new_pop)
        population.increment_evo_step()
        if max_wall_seconds is not None and _time.time() - t_start > max_wall_seconds:
            break

    return population.agents, fitness_history
