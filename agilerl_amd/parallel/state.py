"""Process-group state for one-process-per-GPU training.

First-party replacement for HF Accelerate's launcher plumbing (SURVEY
§1.12 / §5.8): plain ``torch.distributed`` with the nccl backend (which IS
RCCL on ROCm) over the node's xGMI mesh, gloo on CPU-only hosts.  Reads
the standard torchrun env (RANK / LOCAL_RANK / WORLD_SIZE / MASTER_*).
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

__all__ = ["DistributedState", "barrier"]


@dataclass
class DistributedState:
    rank: int = 0
    local_rank: int = 0
    world_size: int = 1
    backend: str = "none"
    device: str = "cpu"

    _instance: Optional["DistributedState"] = None

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    @property
    def is_main(self) -> bool:
        return self.rank == 0

    # ------------------------------------------------------------------
    @classmethod
    def get(cls) -> "DistributedState":
        if cls._instance is None:
            cls._instance = cls._init()
        return cls._instance

    @classmethod
    def _init(cls) -> "DistributedState":
        world_size = int(os.environ.get("WORLD_SIZE", "1"))
        rank = int(os.environ.get("RANK", "0"))
        local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
        if world_size <= 1:
            device = "cuda:0" if torch.cuda.is_available() else "cpu"
            if torch.cuda.is_available():
                torch.cuda.set_device(0)
            return cls(0, 0, 1, "none", device)

        use_gpu = torch.cuda.is_available()
        backend = "nccl" if use_gpu else "gloo"
        if use_gpu:
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        if not dist.is_initialized():
            dist.init_process_group(
                backend=backend,
                rank=rank,
                world_size=world_size,
                timeout=datetime.timedelta(seconds=600),
            )
        device = f"cuda:{local_rank % torch.cuda.device_count()}" if use_gpu else "cpu"
        return cls(rank, local_rank, world_size, backend, device)

    @classmethod
    def reset(cls) -> None:
        cls._instance = None


def barrier() -> None:
    state = DistributedState.get()
    if state.is_distributed and dist.is_initialized():
        if state.backend == "nccl":
            dist.barrier(device_ids=[torch.cuda.current_device()])
        else:
            dist.barrier()


def aggregate_metrics_across_ranks(stats: dict) -> dict:
    """Mean of scalar metrics over all DP ranks (reference
    aggregate_metrics_across_gpus, llm_utils.py:569).  No-op when not
    distributed; non-scalar values pass through from the local rank."""
    state = DistributedState.get()
    if not state.is_distributed:
        return stats
    keys = sorted(k for k, v in stats.items() if isinstance(v, (int, float)) and not isinstance(v, bool))
    if not keys:
        return stats
    t = torch.tensor([float(stats[k]) for k in keys], dtype=torch.float64)
    if state.backend == "nccl":
        t = t.to(state.device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    t = (t / state.world_size).cpu()
    out = dict(stats)
    out.update({k: float(t[i]) for i, k in enumerate(keys)})
    return out
