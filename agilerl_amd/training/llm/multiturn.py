"""Multi-turn LLM fine-tune loop.

Reference parity: ``agilerl/training/llm/multiturn.py:43``
(finetune_llm_multiturn over SyncMultiTurnVecEnv): per episode batch the
agent generates turn-by-turn, the env stitches feedback into the next
prompt, and the padded trajectories train GRPO-family objectives with
trajectory-level rewards.
"""

from __future__ import annotations

import time as _time
from typing import List, Optional

import numpy as np

from ...hpo.mutation import Mutations
from ...hpo.tournament import TournamentSelection
from ...population import Population

__all__ = ["finetune_llm_multiturn"]


def rollout_multiturn(agent, env):
    """One synchronized multi-turn episode batch -> trajectory experiences.

    When the agent generates through the paged engine it exposes the
    behavior-policy sampling logprobs of the completions
    (``last_sampling_logps``); they thread into the trajectories for the
    truncated-IS correction (reference collect_rollouts_llm ->
    sync_vec_env.step(completion_ids, sampling_logps))."""
    prompts = env.reset()
    done = False
    while not done:
        sequences = agent.get_action(prompts, training=True)
        samp = getattr(agent, "last_sampling_logps", None)
        try:
            prompts, done = env.step(sequences, sampling_logps=samp)
        except TypeError:  # env without sampling-logp support
            prompts, done = env.step(sequences)
    return env.get_trajectories()


def finetune_llm_multiturn(
    env,
    pop: List,
    max_steps: int = 1000,
    evo_steps: int = 20,
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
            stats_acc = []
            for _ in range(evo_steps):
                experiences = rollout_multiturn(agent, env)
                stats_acc.append(agent.learn(experiences))
                agent.steps[-1] += 1
            metrics.finalize_training_step(evo_steps)
            for key in stats_acc[0]:
                metrics.log(key, float(np.mean([s[key] for s in stats_acc])))
            # fitness: mean trajectory reward of a greedy episode batch
            rewards = []
            for _ in range(eval_loop):
                prompts = env.reset()
                done = False
                while not done:
                    sequences = agent.get_action(prompts, training=False)
                    prompts, done = env.step(sequences)
                rewards.append(float(env.get_trajectories()["rewards"].mean()))
            fitness = float(np.mean(rewards))
            agent.fitness.append(fitness)
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
