"""Multi-agent off-policy population loop (MADDPG / MATD3).

Reference parity: ``agilerl/training/train_multi_agent_off_policy.py``.
"""

from __future__ import annotations

import time as _time
from typing import List, Optional

import numpy as np

from ..components.replay_buffer import ReplayBuffer
from ..hpo.mutation import Mutations
from ..hpo.tournament import TournamentSelection
from ..population import Population
from .train_off_policy import save_population_checkpoint

__all__ = ["train_multi_agent_off_policy"]


def train_multi_agent_off_policy(
    env,
    env_name: str,
    algo: str,
    pop: List,
    memory: ReplayBuffer,
    max_steps: int = 50_000,
    evo_steps: int = 5_000,
    eval_steps: Optional[int] = None,
    eval_loop: int = 1,
    learning_delay: int = 0,
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
    num_envs = env.num_envs
    agent_ids = pop[0].agent_ids
    fitness_history: List[List[float]] = []
    last_checkpoint = 0

    while population.all_below(max_steps) and not population.should_stop(target):
        for agent, metrics in zip(population.agents, population.metrics):
            metrics.init_training_step()
            obs, _ = env.reset()
            steps_this_cycle = 0
            losses = []
            iters = max(evo_steps // num_envs, 1)
            for it in range(iters):
                env_actions, raw_actions = agent.get_action(obs, training=True)
                next_obs, rewards, term, trunc, info = env.step(env_actions)
                store_next = next_obs
                done_any = np.any([term[a] | trunc[a] for a in agent_ids], axis=0)
                if done_any.any() and "final_observation" in info:
                    store_next = {a: next_obs[a].copy() for a in agent_ids}
                    for a in agent_ids:
                        store_next[a][done_any] = info["final_observation"][a][done_any]
                memory.add(
                    obs=obs,
                    action=raw_actions,
                    reward={a: rewards[a].astype(np.float32) for a in agent_ids},
                    next_obs=store_next,
                    done={a: term[a].astype(np.float32) for a in agent_ids},
                )
                obs = next_obs
                steps_this_cycle += num_envs
                if (
                    len(memory) >= max(agent.batch_size, learning_delay)
                    and it % agent.learn_step == 0
                ):
                    losses.append(agent.learn(memory.sample(agent.batch_size)))
            agent.steps[-1] += steps_this_cycle
            metrics.finalize_training_step(steps_this_cycle)
            if losses:
                metrics.log("loss", float(np.mean(losses)))
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
        if checkpoint is not None and checkpoint_path is not None:
            if population.global_step - last_checkpoint >= checkpoint:
                save_population_checkpoint(population.agents, checkpoint_path)
                last_checkpoint = population.global_step
        if max_wall_seconds is not None and _time.time() - t_start > max_wall_seconds:
            break

    return population.agents, fitness_history
