"""Metric sinks: stdout / CSV / TensorBoard / (optional) wandb.

Reference parity: ``agilerl/logger.py`` (StdOutLogger :87, WandbLogger
:124, CSVLogger :170, TensorboardLogger :210).
"""

from __future__ import annotations

import csv
import os
import warnings
import sys
from typing import Any, Dict, List, Optional

__all__ = ["Logger", "StdOutLogger", "CSVLogger", "TensorboardLogger", "WandbLogger", "PrometheusLogger", "make_loggers"]


class Logger:
    def log_report(self, report: Dict[str, Any]) -> None:  # pragma: no cover
        raise NotImplementedError

    def close(self) -> None:
        pass


class StdOutLogger(Logger):
    def __init__(self, print_every: int = 1):
        self.print_every = print_every
        self._count = 0

    def log_report(self, report: Dict[str, Any]) -> None:
        self._count += 1
        if self._count % self.print_every:
            return
        pop = report.get("population", [])
        step = report.get("global_step", 0)
        fps = report.get("mean_steps_per_sec", 0.0)
        best = report.get("best_fitness", float("nan"))
        mean = report.get("mean_fitness", float("nan"))
        line = (
            f"[step {step:>9}] fps={fps:9.1f} best_fitness={best:10.2f} "
            f"mean_fitness={mean:10.2f} pop={len(pop)}"
        )
        muts = [a.get("mut", "-") for a in pop]
        if any(m not in ("None", "-") for m in muts):
            line += f" muts={muts}"
        print(line, flush=True)


class CSVLogger(Logger):
    def __init__(self, path: str = "training_log.csv"):
        self.path = path
        self._writer = None
        self._file = None
        self._fields: Optional[List[str]] = None

    def log_report(self, report: Dict[str, Any]) -> None:
        rows = []
        base = {k: v for k, v in report.items() if not isinstance(v, (list, dict))}
        for agent_snap in report.get("population", [{}]):
            row = dict(base)
            row.update({k: v for k, v in agent_snap.items() if not isinstance(v, (list, dict))})
            rows.append(row)
        for row in rows:
            if self._writer is None:
                self._fields = sorted(row.keys())
                self._file = open(self.path, "w", newline="")
                self._writer = csv.DictWriter(self._file, fieldnames=self._fields, extrasaction="ignore")
                self._writer.writeheader()
            self._writer.writerow(row)
        if self._file:
            self._file.flush()

    def close(self) -> None:
        if self._file:
            self._file.close()


class PrometheusLogger(Logger):
    """Exposes population metrics as Prometheus gauges on an HTTP endpoint.

    Production-serving observability sink (prometheus_client is a hard
    dependency of the serving stack, optional here — degrades to a no-op
    with a warning when unavailable).  Scalar report fields become
    ``agilerl_<name>`` gauges; per-agent scalars become
    ``agilerl_agent_<name>{agent="<idx>"}``.
    """

    def __init__(self, port: int = 9300, addr: str = "127.0.0.1", start_server: bool = True):
        try:
            from prometheus_client import Gauge, start_http_server
        except ImportError:  # pragma: no cover
            warnings.warn("prometheus_client not installed; PrometheusLogger disabled")
            self._gauge_cls = None
            return
        self._gauge_cls = Gauge
        self._gauges = {}
        self._agent_gauges = {}
        if start_server:
            start_http_server(port, addr=addr)

    @staticmethod
    def _sanitize(name: str) -> str:
        return "".join(c if c.isalnum() or c == "_" else "_" for c in name)

    def log_report(self, report: Dict[str, Any]) -> None:
        if self._gauge_cls is None:
            return
        for key, val in report.items():
            if isinstance(val, bool) or not isinstance(val, (int, float)):
                continue
            name = "agilerl_" + self._sanitize(key)
            if name not in self._gauges:
                self._gauges[name] = self._gauge_cls(name, key)
            self._gauges[name].set(float(val))
        for snap in report.get("population", []):
            idx = str(snap.get("index", "?"))
            for key, val in snap.items():
                if isinstance(val, bool) or not isinstance(val, (int, float)):
                    continue
                name = "agilerl_agent_" + self._sanitize(key)
                if name not in self._agent_gauges:
                    self._agent_gauges[name] = self._gauge_cls(name, key, ["agent"])
                self._agent_gauges[name].labels(agent=idx).set(float(val))


class TensorboardLogger(Logger):
    def __init__(self, log_dir: str = "runs"):
        try:
            from torch.utils.tensorboard import SummaryWriter

            self.writer = SummaryWriter(log_dir)
        except ImportError:
            self.writer = None
            print("tensorboard not installed; TensorboardLogger disabled", file=sys.stderr)

    def log_report(self, report: Dict[str, Any]) -> None:
        if self.writer is None:
            return
        step = report.get("global_step", 0)
        for k, v in report.items():
            if isinstance(v, (int, float)):
                self.writer.add_scalar(f"population/{k}", v, step)
        for snap in report.get("population", []):
            idx = snap.get("agent", 0)
            for k, v in snap.items():
                if isinstance(v, (int, float)):
                    self.writer.add_scalar(f"agent_{idx}/{k}", v, step)

    def close(self) -> None:
        if self.writer is not None:
            self.writer.close()


class WandbLogger(Logger):
    def __init__(self, project: str = "agilerl-amd", **init_kwargs):
        try:
            import wandb  # noqa: F401

            self.wandb = wandb
            self.run = wandb.init(project=project, **init_kwargs)
        except ImportError:
            self.wandb = None
            self.run = None
            print("wandb not installed; WandbLogger disabled", file=sys.stderr)

    def log_report(self, report: Dict[str, Any]) -> None:
        if self.run is None:
            return
        flat = {k: v for k, v in report.items() if isinstance(v, (int, float))}
        for snap in report.get("population", []):
            idx = snap.get("agent", 0)
            flat.update(
                {f"agent_{idx}/{k}": v for k, v in snap.items() if isinstance(v, (int, float))}
            )
        self.wandb.log(flat, step=report.get("global_step", 0))

    def close(self) -> None:
        if self.run is not None:
            self.run.finish()


def make_loggers(
    stdout: bool = True,
    csv_path: Optional[str] = None,
    tensorboard_dir: Optional[str] = None,
    wandb_project: Optional[str] = None,
    prometheus_port: Optional[int] = None,
) -> List[Logger]:
    loggers: List[Logger] = []
    if stdout:
        loggers.append(StdOutLogger())
    if csv_path:
        loggers.append(CSVLogger(csv_path))
    if tensorboard_dir:
        loggers.append(TensorboardLogger(tensorboard_dir))
    if wandb_project:
        loggers.append(WandbLogger(wandb_project))
    if prometheus_port:
        loggers.append(PrometheusLogger(port=prometheus_port))
    return loggers
