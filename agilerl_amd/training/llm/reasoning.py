"""Single-turn LLM reasoning fine-tune loop (GRPO family).

Reference parity: ``agilerl/training/llm/reasoning.py:42``
(finetune_llm_reasoning): epochs of {generate -> reward -> learn}, with
tournament selection + RL-HP mutations every ``evo_steps`` and adapter
checkpointing.
"""

from __future__ import annotations

import time as _time
from typing import List, Optional

import numpy as np

from ...hpo.mutation import Mutations
from ...hpo.tournament import TournamentSelection
from ...llm_envs.base import make_grpo_experiences
from ...population import Population

__all__ = ["finetune_llm_reasoning"]


def finetune_llm_reasoning(
    env,
    pop: List,
    max_steps: int = 1000,
    evo_steps: int = 50,
    eval_loop: int = 1,
    target: Optional[float] = None,
    tournament: Optional[TournamentSelection] = None,
    mutation: Optional[Mutations] = None,
    checkpoint_steps: Optional[int] = None,
    checkpoint_path: Optional[str] = None,
    loggers: Optional[List] = None,
    max_wall_seconds: Optional[float] = None,
    verbose: bool = True,
):
    """max_steps counts prompt-batches ("episodes") per agent."""
    t_start = _time.time()
    if loggers is None and verbose:
        from ...logger import StdOutLogger

        loggers = [StdOutLogger()]
    population = Population(pop, loggers or [])
    pad_id = getattr(pop[0].model.config, "pad_token_id", None) or 0
    fitness_history: List[List[float]] = []
    last_ckpt = 0

    while population.all_below(max_steps) and not population.should_stop(target):
        for agent, metrics in zip(population.agents, population.metrics):
            metrics.init_training_step()
            stats_acc = []
            for _ in range(evo_steps):
                prompts = env.reset()
                sequences = agent.get_action(prompts, training=True)
                rewards = env.score(sequences)
                experiences = make_grpo_experiences(env, sequences, rewards, pad_token_id=pad_id)
                stats = agent.learn(experiences)
                stats_acc.append(stats)
                agent.steps[-1] += 1
            metrics.finalize_training_step(evo_steps)
            mean_stats = {k: float(np.mean([s[k] for s in stats_acc])) for k in stats_acc[0]}
            from ...parallel import aggregate_metrics_across_ranks

            for key, val in aggregate_metrics_across_ranks(mean_stats).items():
                metrics.log(key, val)
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
            for agent in old:  # free replaced adapter slots in the shared model
                if agent not in new_pop:
                    agent.clean_up()
            population.replace(new_pop)
        population.increment_evo_step()
        if checkpoint_steps is not None and checkpoint_path is not None:
            step = population.global_step
            if step - last_ckpt >= checkpoint_steps:
                for i, agent in enumerate(population.agents):
                    agent.save_checkpoint(f"{checkpoint_path}_{i}")
                last_ckpt = step
        if max_wall_seconds is not None and _time.time() - t_start > max_wall_seconds:
            break

    return population.agents, fitness_history
