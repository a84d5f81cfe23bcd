"""Population container + metrics aggregation.

Reference parity: ``agilerl/population.py`` (Population :474,
PopulationMetrics :72, MetricsReport :212) — owns the logger pipeline,
aggregates per-agent metrics, handles the stop condition.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import numpy as np

from .logger import Logger
from .metrics import AgentMetrics

__all__ = ["Population"]


class Population:
    def __init__(self, agents: List, loggers: Optional[List[Logger]] = None):
        self.agents = list(agents)
        for i, agent in enumerate(self.agents):
            agent.index = i
        self.metrics: List[AgentMetrics] = [AgentMetrics(i) for i in range(len(self.agents))]
        self.loggers = loggers or []
        self.evo_step = 0

    def __len__(self) -> int:
        return len(self.agents)

    def __iter__(self):
        return iter(self.agents)

    def __getitem__(self, idx: int):
        return self.agents[idx]

    # ------------------------------------------------------------------
    @property
    def global_step(self) -> int:
        return int(sum(a.steps[-1] for a in self.agents))

    def all_below(self, max_steps: int) -> bool:
        return all(a.steps[-1] < max_steps for a in self.agents)

    def should_stop(self, target: Optional[float]) -> bool:
        if target is None:
            return False
        return any(
            a.fitness and np.mean(a.fitness[-3:]) >= target for a in self.agents
        )

    @property
    def best_agent(self):
        fits = [a.fitness[-1] if a.fitness else -np.inf for a in self.agents]
        return self.agents[int(np.argmax(fits))]

    # ------------------------------------------------------------------
    def replace(self, new_agents: List) -> None:
        """Swap in the post-selection population, keeping metric continuity."""
        self.agents = list(new_agents)
        for i, agent in enumerate(self.agents):
            agent.index = i
        old = self.metrics
        self.metrics = []
        for i, agent in enumerate(self.agents):
            m = AgentMetrics(i)
            # carry global step so steps/sec stays meaningful
            src = old[i] if i < len(old) else None
            if src is not None:
                m.global_step = src.global_step
                m.fitness_window = src.fitness_window
            self.metrics.append(m)

    def increment_evo_step(self) -> None:
        self.evo_step += 1

    # ------------------------------------------------------------------
    def build_report(self) -> Dict[str, Any]:
        snaps = [m.snapshot(a) for m, a in zip(self.metrics, self.agents)]
        fits = [a.fitness[-1] for a in self.agents if a.fitness]
        report: Dict[str, Any] = {
            "global_step": self.global_step,
            "evo_step": self.evo_step,
            "mean_steps_per_sec": float(np.sum([m.steps_per_sec for m in self.metrics])),
            "best_fitness": float(np.max(fits)) if fits else float("nan"),
            "mean_fitness": float(np.mean(fits)) if fits else float("nan"),
            "population": snaps,
        }
        return report

    def report_metrics(self) -> Dict[str, Any]:
        report = self.build_report()
        for logger in self.loggers:
            logger.log_report(report)
        for m in self.metrics:
            m.clear_cycle()
        return report
