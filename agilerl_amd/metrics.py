"""Per-agent metric accumulators.

Reference parity: ``agilerl/metrics.py`` (AgentMetrics :194,
MultiAgentMetrics :291) — scalar accumulators, histogram deques,
hyperparameter snapshots, steps/sec timing (the BASELINE headline metric)
and a fitness window.
"""

from __future__ import annotations

import time
from collections import deque
from typing import Any, Deque, Dict, List, Optional

import numpy as np

__all__ = ["AgentMetrics", "MultiAgentMetrics"]


class AgentMetrics:
    def __init__(self, agent_index: int = 0, fitness_window: int = 5, histogram_len: int = 100):
        self.agent_index = agent_index
        self.scalars: Dict[str, List[float]] = {}
        self.histograms: Dict[str, Deque[float]] = {}
        self.histogram_len = histogram_len
        self.fitness_window: Deque[float] = deque(maxlen=fitness_window)
        self.global_step = 0
        self._step_t0: Optional[float] = None
        self._step_count_at_t0 = 0
        self.steps_per_sec = 0.0

    # ------------------------------------------------------------------
    def log(self, name: str, value: float) -> None:
        self.scalars.setdefault(name, []).append(float(value))

    def log_histogram(self, name: str, value: float) -> None:
        self.histograms.setdefault(name, deque(maxlen=self.histogram_len)).append(float(value))

    def log_fitness(self, fitness: float) -> None:
        self.fitness_window.append(float(fitness))
        self.log("fitness", fitness)

    # ------------------------------------------------------------------
    def init_training_step(self) -> None:
        self._step_t0 = time.perf_counter()
        self._step_count_at_t0 = self.global_step

    def finalize_training_step(self, steps_done: int) -> None:
        self.global_step += steps_done
        if self._step_t0 is not None:
            dt = time.perf_counter() - self._step_t0
            if dt > 0:
                self.steps_per_sec = (self.global_step - self._step_count_at_t0) / dt

    # ------------------------------------------------------------------
    @property
    def mean_fitness(self) -> float:
        return float(np.mean(self.fitness_window)) if self.fitness_window else float("-inf")

    def snapshot(self, agent=None) -> Dict[str, Any]:
        out: Dict[str, Any] = {
            "agent": self.agent_index,
            "global_step": self.global_step,
            "steps_per_sec": self.steps_per_sec,
            "mean_fitness": self.mean_fitness,
        }
        for name, vals in self.scalars.items():
            if vals:
                out[name] = float(np.mean(vals[-20:]))
        if agent is not None:
            out["mut"] = getattr(agent, "mut", "None")
            for hp in getattr(agent, "hp_config").names() if hasattr(agent, "hp_config") else []:
                out[f"hp/{hp}"] = getattr(agent, hp, None)
        return out

    def clear_cycle(self) -> None:
        self.scalars.clear()


class MultiAgentMetrics(AgentMetrics):
    """Adds per-sub-agent score tracking for multi-agent algorithms."""

    def __init__(self, agent_index: int = 0, agent_ids: Optional[List[str]] = None, **kwargs):
        super().__init__(agent_index, **kwargs)
        self.agent_ids = agent_ids or []

    def log_agent(self, agent_id: str, name: str, value: float) -> None:
        self.log(f"{agent_id}/{name}", value)
