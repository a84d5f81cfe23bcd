"""Multi-agent on-policy population loop (IPPO).

Reference parity: ``agilerl/training/train_multi_agent_on_policy.py``.
"""

from __future__ import annotations

import time as _time
from typing import Dict, List, Optional

import numpy as np

from ..components.rollout_buffer import RolloutBuffer
from ..hpo.mutation import Mutations
from ..hpo.tournament import TournamentSelection
from ..population import Population
from .train_off_policy import save_population_checkpoint

__all__ = ["train_multi_agent_on_policy"]


def _collect_ma_rollout(agent, env, buffers: Dict[str, RolloutBuffer], n_steps: int, obs):
    agent_ids = agent.agent_ids
    if obs is None:
        obs, _ = env.reset()
    for b in buffers.values():
        b.reset()
    for _ in range(n_steps):
        env_actions, log_probs, values = agent.get_action(obs, training=True)
        next_obs, rewards, term, trunc, info = env.step(env_actions)
        for aid in agent_ids:
            done = (term[aid] | trunc[aid]).astype(np.float32)
            buffers[aid].add(
                obs=obs[aid],
                action=env_actions[aid],
                reward=rewards[aid],
                done=done,
                value=values[aid],
                log_prob=log_probs[aid],
            )
        obs = next_obs
    last_values = agent.get_values(obs)
    for aid in agent_ids:
        buffers[aid].compute_returns_and_advantages(last_values[aid])
    return obs


def train_multi_agent_on_policy(
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
    fitness_history: List[List[float]] = []
    last_checkpoint = 0

    def make_buffers(agent):
        return {
            aid: RolloutBuffer(
                capacity=agent.learn_step, num_envs=num_envs, device=agent.device,
                gamma=agent.gamma, gae_lambda=agent.gae_lambda,
            )
            for aid in agent.agent_ids
        }

    buffer_map = {id(a): make_buffers(a) for a in population.agents}

    while population.all_below(max_steps) and not population.should_stop(target):
        for agent, metrics in zip(population.agents, population.metrics):
            metrics.init_training_step()
            buffers = buffer_map.get(id(agent)) or make_buffers(agent)
            buffer_map[id(agent)] = buffers
            steps_this_cycle = 0
            obs = None
            learn_stats = []
            while steps_this_cycle < evo_steps:
                obs = _collect_ma_rollout(agent, env, buffers, agent.learn_step, obs)
                learn_stats.append(agent.learn(buffers))
                steps_this_cycle += agent.learn_step * num_envs
            agent.steps[-1] += steps_this_cycle
            metrics.finalize_training_step(steps_this_cycle)
            for key in ("policy_loss", "value_loss", "entropy"):
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
            new_pop = mutation.mutation(new_pop)
            buffer_map = {id(a): make_buffers(a) for a in new_pop}
            population.replace(new_pop)
        population.increment_evo_step()
        if checkpoint is not None and checkpoint_path is not None:
            if population.global_step - last_checkpoint >= checkpoint:
                save_population_checkpoint(population.agents, checkpoint_path)
                last_checkpoint = population.global_step
        if max_wall_seconds is not None and _time.time() - t_start > max_wall_seconds:
            break

    return population.agents, fitness_history
