"""Offline RL population loop (CQN on fixed datasets).

Reference parity: ``agilerl/training/train_offline.py`` (minari datasets;
here datasets are dicts/npz of transitions — no hub access offline).
"""

from __future__ import annotations

import time as _time
from typing import Dict, List, Optional

import numpy as np

from ..components.replay_buffer import ReplayBuffer
from ..hpo.mutation import Mutations
from ..hpo.tournament import TournamentSelection
from ..population import Population
from .train_off_policy import save_population_checkpoint

__all__ = ["train_offline", "load_transitions_into_buffer"]


def load_transitions_into_buffer(dataset: Dict[str, np.ndarray], memory: ReplayBuffer,
                                 chunk: int = 4096) -> None:
    """dataset keys: observations, actions, rewards, next_observations,
    terminals (minari-style arrays)."""
    n = len(dataset["observations"])
    for s in range(0, n, chunk):
        e = min(s + chunk, n)
        memory.add(
            obs=dataset["observations"][s:e],
            action=dataset["actions"][s:e],
            reward=np.asarray(dataset["rewards"][s:e], dtype=np.float32),
            next_obs=dataset["next_observations"][s:e],
            done=np.asarray(dataset["terminals"][s:e], dtype=np.float32),
        )


def train_offline(
    env,
    env_name: str,
    dataset: Dict[str, np.ndarray],
    algo: str,
    pop: List,
    memory: Optional[ReplayBuffer] = None,
    max_steps: int = 50_000,
    evo_steps: int = 5_000,
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
    """max_steps counts gradient steps per agent (no env interaction)."""
    t_start = _time.time()
    if loggers is None and verbose:
        from ..logger import StdOutLogger

        loggers = [StdOutLogger()]
    population = Population(pop, loggers or [])
    if memory is None:
        memory = ReplayBuffer(len(dataset["observations"]))
    if len(memory) == 0:
        load_transitions_into_buffer(dataset, memory)
    fitness_history: List[List[float]] = []
    last_checkpoint = 0

    while population.all_below(max_steps) and not population.should_stop(target):
        for agent, metrics in zip(population.agents, population.metrics):
            metrics.init_training_step()
            losses = []
            for _ in range(evo_steps):
                losses.append(agent.learn(memory.sample(agent.batch_size)))
            agent.steps[-1] += evo_steps
            metrics.finalize_training_step(evo_steps)
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


def save_transitions(dataset: Dict[str, np.ndarray], path: str) -> None:
    """Persist a transition dataset as compressed npz (offline-RL format)."""
    np.savez_compressed(path, **dataset)


def load_transitions(path: str) -> Dict[str, np.ndarray]:
    with np.load(path) as z:
        return {k: z[k] for k in z.files}


def collect_transitions(env, agent=None, steps: int = 1000,
                        epsilon: float = 1.0) -> Dict[str, np.ndarray]:
    """Roll a (possibly random) policy to build an offline dataset.

    agent=None or epsilon=1.0 gives uniform-random behavior; otherwise
    epsilon-greedy around ``agent.get_action``.  Returns minari-style
    arrays consumable by load_transitions_into_buffer / save_transitions.
    """
    obs_l, act_l, rew_l, next_l, term_l = [], [], [], [], []
    obs, _ = env.reset()
    for _ in range(steps):
        if agent is None or np.random.rand() < epsilon:
            action = np.array(
                [env.action_space.sample() for _ in range(env.num_envs)]
            ) if hasattr(env.action_space, "sample") else np.random.randint(
                0, env.action_space.n, env.num_envs
            )
        else:
            action = agent.get_action(obs, training=False)
        next_obs, reward, term, trunc, _ = env.step(action)
        obs_l.append(np.asarray(obs))
        act_l.append(np.asarray(action))
        rew_l.append(np.asarray(reward, dtype=np.float32))
        next_l.append(np.asarray(next_obs))
        term_l.append(np.asarray(term, dtype=np.float32))
        obs = next_obs
    return {
        "observations": np.concatenate(obs_l),
        "actions": np.concatenate(act_l),
        "rewards": np.concatenate(rew_l),
        "next_observations": np.concatenate(next_l),
        "terminals": np.concatenate(term_l),
    }
