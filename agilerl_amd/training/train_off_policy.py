"""Off-policy population training loop (DQN / Rainbow / DDPG / TD3).

Reference parity: ``agilerl/training/train_off_policy.py:121``
(collect -> learn -> eval -> evolve -> checkpoint per evo cycle).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import numpy as np

from ..components.replay_buffer import PrioritizedReplayBuffer, ReplayBuffer
from ..hpo.mutation import Mutations
from ..hpo.tournament import TournamentSelection
from ..population import Population

__all__ = ["train_off_policy"]


def _merge_final_obs(next_obs, final_obs, mask):
    """Bootstrap targets use the pre-reset observation for done rows;
    handles Dict/Tuple observation structures."""
    if isinstance(next_obs, dict):
        out = {k: v.copy() for k, v in next_obs.items()}
        for k in out:
            out[k][mask] = final_obs[k][mask]
        return out
    if isinstance(next_obs, (tuple, list)):
        out = [v.copy() for v in next_obs]
        for o, f in zip(out, final_obs):
            o[mask] = f[mask]
        return type(next_obs)(out)
    out = next_obs.copy()
    out[mask] = final_obs[mask]
    return out


def train_off_policy(
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
    eps_start: float = 1.0,
    eps_end: float = 0.05,
    eps_decay: float = 0.995,
    per_beta_start: float = 0.4,
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
    """Train a population of off-policy agents with evolutionary HPO.

    Returns (population_agents, fitness_history).
    """
    import time as _time

    t_start = _time.time()
    if loggers is None and verbose:
        from ..logger import StdOutLogger

        loggers = [StdOutLogger()]
    population = Population(pop, loggers or [])
    per = isinstance(memory, PrioritizedReplayBuffer)
    eps = eps_start
    num_envs = env.num_envs
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
                action = agent.get_action(obs, epsilon=eps)
                next_obs, reward, term, trunc, info = env.step(action)
                store_next = next_obs
                done_any = term | trunc
                if done_any.any() and "final_observation" in info:
                    store_next = _merge_final_obs(next_obs, info["final_observation"], done_any)
                memory.add(
                    obs=obs,
                    action=action,
                    reward=reward,
                    next_obs=store_next,
                    done=term.astype(np.float32),
                )
                obs = next_obs
                eps = max(eps_end, eps * eps_decay)
                steps_this_cycle += num_envs

                total_steps = population.global_step + steps_this_cycle
                frac = min(total_steps / max_steps, 1.0)
                if (
                    len(memory) >= max(agent.batch_size, learning_delay)
                    and it % agent.learn_step == 0
                ):
                    combined = bool(getattr(agent, "combined_reward", False)) and \
                        getattr(memory, "n_step", 1) > 1
                    if per:
                        beta = per_beta_start + (1.0 - per_beta_start) * frac
                        batch = memory.sample(agent.batch_size, beta=beta,
                                              include_one_step=combined)
                    elif combined:
                        batch = memory.sample(agent.batch_size, include_one_step=True)
                    else:
                        batch = memory.sample(agent.batch_size)
                    loss = agent.learn(batch)
                    losses.append(loss)
                    if per and hasattr(agent, "last_td_errors") and agent.last_td_errors is not None:
                        memory.update_priorities(batch["idxs"], agent.last_td_errors)
            agent.steps[-1] += steps_this_cycle
            metrics.finalize_training_step(steps_this_cycle)
            if losses:
                metrics.log("loss", float(np.mean(losses)))
            metrics.log("epsilon", eps)

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


def save_population_checkpoint(
    agents: List, path: str, overwrite: bool = True, step: int = 0
) -> None:
    """One checkpoint file per agent (reference utils/utils.py:1171)."""
    base, ext = os.path.splitext(path)
    ext = ext or ".pt"
    for i, agent in enumerate(agents):
        suffix = f"_{i}" if overwrite else f"_{i}_{step}"
        agent.save_checkpoint(f"{base}{suffix}{ext}")
