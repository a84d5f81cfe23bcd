"""On-policy population training loop (PPO).

Reference parity: ``agilerl/training/train_on_policy.py`` (delegates
stepping to ``rollouts/``, :218-254).
"""

from __future__ import annotations

import time as _time
from typing import List, Optional

import numpy as np

from ..components.rollout_buffer import RolloutBuffer
from ..hpo.mutation import Mutations
from ..hpo.tournament import TournamentSelection
from ..population import Population
from ..rollouts.on_policy import collect_rollouts
from .train_off_policy import save_population_checkpoint

__all__ = ["train_on_policy"]


def train_on_policy(
    env,
    env_name: str,
    algo: str,
    pop: List,
    max_steps: int = 100_000,
    evo_steps: int = 10_000,
    eval_steps: Optional[int] = None,
    eval_loop: int = 1,
    target: Optional[float] = None,
    tournament: Optional[TournamentSelection] = None,
    mutation: Optional[Mutations] = None,
    checkpoint: Optional[int] = None,
    checkpoint_path: Optional[str] = None,
    overwrite_checkpoints: bool = True,
    loggers: Optional[List] = None,
    save_elite: bool = False,
    elite_path: Optional[str] = None,
    max_wall_seconds: Optional[float] = None,
    verbose: bool = True,
):
    t_start = _time.time()
    if loggers is None and verbose:
        from ..logger import StdOutLogger

        loggers = [StdOutLogger()]
    population = Population(pop, loggers or [])
    num_envs = env.num_envs
    fitness_history: List[List[float]] = []
    last_checkpoint = 0

    # one rollout buffer per agent (rollout length = agent.learn_step)
    buffers = {
        id(agent): RolloutBuffer(
            capacity=agent.learn_step,
            num_envs=num_envs,
            device=agent.device,
            gamma=agent.gamma,
            gae_lambda=agent.gae_lambda,
        )
        for agent in population.agents
    }

    while population.all_below(max_steps) and not population.should_stop(target):
        for agent, metrics in zip(population.agents, population.metrics):
            metrics.init_training_step()
            buffer = buffers.get(id(agent))
            if buffer is None or buffer.capacity != agent.learn_step:
                buffer = RolloutBuffer(
                    capacity=agent.learn_step,
                    num_envs=num_envs,
                    device=agent.device,
                    gamma=agent.gamma,
                    gae_lambda=agent.gae_lambda,
                )
                buffers[id(agent)] = buffer
            steps_this_cycle = 0
            obs, done, hidden = None, None, None
            learn_stats = []
            recurrent = getattr(agent, "recurrent", False)
            while steps_this_cycle < evo_steps:
                if recurrent:
                    from ..rollouts.on_policy import collect_rollouts_recurrent

                    obs, done, hidden, _stats = collect_rollouts_recurrent(
                        agent, env, buffer, agent.learn_step, obs, done, hidden
                    )
                else:
                    obs, done, _stats = collect_rollouts(
                        agent, env, buffer, agent.learn_step, obs, done
                    )
                stats = agent.learn(buffer)
                learn_stats.append(stats)
                steps_this_cycle += agent.learn_step * num_envs
            agent.steps[-1] += steps_this_cycle
            metrics.finalize_training_step(steps_this_cycle)
            for key in ("policy_loss", "value_loss", "entropy", "approx_kl"):
                metrics.log(key, float(np.mean([s[key] for s in learn_stats])))

            fitness = agent.test(env, max_steps=eval_steps, loop=eval_loop)
            metrics.log_fitness(fitness)

        fitness_history.append([a.fitness[-1] for a in population.agents])
        if verbose or population.loggers:
            population.report_metrics()

        if population.should_stop(target) or not population.all_below(max_steps):
            break

        if tournament is not None and mutation is not None:
            elite, new_pop = tournament.select(population.agents)
            if save_elite and elite_path:
                elite.save_checkpoint(elite_path)
            new_pop = mutation.mutation(new_pop)
            # old agents were replaced: rebuild buffer map for clones
            for old_agent in list(buffers.keys()):
                pass
            buffers = {
                id(a): RolloutBuffer(
                    capacity=a.learn_step,
                    num_envs=num_envs,
                    device=a.device,
                    gamma=a.gamma,
                    gae_lambda=a.gae_lambda,
                )
                for a in new_pop
            }
            population.replace(new_pop)
        population.increment_evo_step()

        if checkpoint is not None and checkpoint_path is not None:
            if population.global_step - last_checkpoint >= checkpoint:
                save_population_checkpoint(
                    population.agents, checkpoint_path, overwrite=overwrite_checkpoints,
                    step=population.global_step,
                )
                last_checkpoint = population.global_step

        if max_wall_seconds is not None and _time.time() - t_start > max_wall_seconds:
            break

    return population.agents, fitness_history
