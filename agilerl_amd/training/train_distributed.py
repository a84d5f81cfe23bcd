"""Distributed one-agent-per-GPU population training (on-policy).

The headline MI355X redesign (SURVEY §2.2 "population parallelism"):
population slots shard round-robin over ranks (one process per GPU over
RCCL/xGMI), each agent trains wholly on its owner rank, and evolution
rounds run as fitness all-gather + rank-0 tournament plan broadcast +
winner weight transfer (``parallel.DistributedPopulation``).  Degrades to
the sequential single-process mode at world_size=1 — the same code path
produces the 1/2/4/8-GPU scaling curve.

On GPU the collect loop uses the hipGraph-captured collector when the
env supports it (TorchVecEnv + discrete PPO).
"""

from __future__ import annotations

import time as _time
from typing import Callable, Dict, List, Optional

import numpy as np
import torch

from ..components.rollout_buffer import RolloutBuffer
from ..hpo.mutation import Mutations
from ..hpo.tournament import TournamentSelection
from ..parallel import DistributedPopulation, DistributedState
from ..rollouts.on_policy import collect_rollouts, collect_rollouts_device

__all__ = ["train_on_policy_distributed"]


def train_on_policy_distributed(
    agent_factory: Callable[[int], object],
    env_factory: Callable[[int], object],
    pop_size: int = 8,
    max_steps: int = 1_000_000,
    evo_steps: int = 65_536,
    target: Optional[float] = None,
    tournament: Optional[TournamentSelection] = None,
    mutation: Optional[Mutations] = None,
    fitness_window: int = 5,
    use_graph: Optional[bool] = None,
    max_wall_seconds: Optional[float] = None,
    verbose: bool = True,
):
    """Returns (local_agents_by_slot, fitness_history as seen on this rank).

    ``agent_factory(slot)`` / ``env_factory(slot)`` build the agent/env for
    a population slot on the calling rank's device.  Fitness is the running
    mean episodic return from training rollouts (no separate eval pass —
    every env step counts toward max_steps).
    """
    t_start = _time.time()
    state = DistributedState.get()
    pop = DistributedPopulation(agent_factory, pop_size)
    if use_graph is None:
        use_graph = torch.cuda.is_available()

    envs: Dict[int, object] = {}
    buffers: Dict[int, RolloutBuffer] = {}
    collectors: Dict[int, object] = {}
    carried: Dict[int, tuple] = {}
    fit_windows: Dict[int, List[float]] = {}

    def init_slot(slot: int) -> None:
        envs[slot] = env_factory(slot)
        agent = pop.agents[slot]
        buffers[slot] = RolloutBuffer(
            capacity=agent.learn_step, num_envs=envs[slot].num_envs,
            device=agent.device, gamma=agent.gamma, gae_lambda=agent.gae_lambda,
        )
        carried[slot] = (None, None)
        fit_windows.setdefault(slot, [])
        collectors.pop(slot, None)
        if use_graph and getattr(envs[slot], "is_torch", False):
            from ..rollouts.graph_collector import GraphedPPOCollector

            collectors[slot] = GraphedPPOCollector(agent, envs[slot], agent.learn_step)

    for slot in pop.local_indices:
        init_slot(slot)

    fitness_history: List[np.ndarray] = []
    while True:
        # one population cycle: every local slot collects+learns evo_steps
        for slot in pop.local_indices:
            agent = pop.agents[slot]
            env = envs[slot]
            steps_done = 0
            stats = {}
            while steps_done < evo_steps:
                if slot in collectors:
                    col = collectors[slot]
                    if col.agent is not agent or col.n_steps != agent.learn_step:
                        from ..rollouts.graph_collector import GraphedPPOCollector

                        col = GraphedPPOCollector(agent, env, agent.learn_step)
                        collectors[slot] = col
                    flat, stats = col.collect()
                    agent.learn(flat)
                else:
                    buffer = buffers[slot]
                    if buffer.capacity != agent.learn_step:
                        buffer = RolloutBuffer(
                            capacity=agent.learn_step, num_envs=env.num_envs,
                            device=agent.device, gamma=agent.gamma,
                            gae_lambda=agent.gae_lambda,
                        )
                        buffers[slot] = buffer
                    obs, done = carried[slot]
                    collect = (
                        collect_rollouts_device
                        if getattr(env, "is_torch", False)
                        else collect_rollouts
                    )
                    obs, done, stats = collect(agent, env, buffer, agent.learn_step, obs, done)
                    carried[slot] = (obs, done)
                    agent.learn(buffer)
                n = agent.learn_step * env.num_envs
                agent.steps[-1] += n
                steps_done += n
                if "mean_episode_return" in stats:
                    w = fit_windows[slot]
                    w.append(stats["mean_episode_return"])
                    del w[:-fitness_window]
                    agent.fitness.append(float(np.mean(w)))

        fitnesses = pop.gather_fitness()
        fitness_history.append(fitnesses)
        if state.is_main and verbose:
            best = float(np.max(fitnesses))
            print(
                f"[evo {pop.evo_step:>4}] best={best:9.2f} "
                f"mean={float(np.mean(fitnesses)):9.2f} fitnesses={np.round(fitnesses, 1)}",
                flush=True,
            )

        stop = (
            (target is not None and float(np.max(fitnesses)) >= target)
            or pop.global_step >= max_steps * pop_size
            or (max_wall_seconds is not None and _time.time() - t_start > max_wall_seconds)
        )
        if state.is_distributed:
            import torch.distributed as dist

            flag = torch.tensor([1.0 if stop else 0.0])
            if state.backend == "nccl":
                flag = flag.to(state.device)
            dist.all_reduce(flag, op=dist.ReduceOp.MAX)
            stop = bool(flag.item() > 0)
        if stop:
            break

        if tournament is not None:
            from ..parallel.population_runtime import adopt_agent_state

            old_agents = dict(pop.agents)
            pop.evolve(tournament, mutation)
            for slot in pop.local_indices:
                old = old_agents.get(slot)
                new = pop.agents[slot]
                if old is not None and old is not new and adopt_agent_state(old, new):
                    # same architecture: keep the old object so its captured
                    # hipGraphs (collect + learn) survive the evolution round
                    pop.agents[slot] = old
                    continue
                if slot not in envs:
                    init_slot(slot)
                else:
                    carried[slot] = (None, None)
                    col = collectors.get(slot)
                    if col is not None and col.agent is not pop.agents[slot]:
                        init_slot(slot)

    return pop.agents, fitness_history
