"""First-party data-parallel gradient synchronization over RCCL.

No Accelerate/DeepSpeed: a thin bucketed-allreduce layer for the
"one agent spans all GPUs" mode (LLM fine-tuning).

MI355X sizing: xGMI is 7 point-to-point links x ~153 GB/s per GPU; ring
collectives are per-link bound, so buckets are large (64 MB default) to
amortize launch/latency and let RCCL stripe multi-ring across links.
Overlap: gradient hooks fire as backward produces grads; buckets launch
async allreduces on the communication stream and ``finalize()`` waits
before ``optimizer.step()``.
"""

from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist

from .state import DistributedState

__all__ = ["GradBucketer", "wrap_ddp", "allreduce_gradients", "broadcast_module"]


class GradBucketer:
    """Bucketed async gradient all-reduce, attached via post-accumulate hooks."""

    def __init__(
        self,
        module: torch.nn.Module,
        process_group=None,
        bucket_bytes: int = 64 << 20,
    ):
        self.module = module
        self.group = process_group
        self.bucket_bytes = bucket_bytes
        self.state = DistributedState.get()
        self._params: List[torch.nn.Parameter] = [
            p for p in module.parameters() if p.requires_grad
        ]
        self._pending: List[torch.Tensor] = []
        self._pending_bytes = 0
        self._works: List = []
        self._hooks = []
        if self.state.is_distributed:
            for p in self._params:
                h = p.register_post_accumulate_grad_hook(self._on_grad)
                self._hooks.append(h)

    def _on_grad(self, param: torch.nn.Parameter) -> None:
        if param.grad is None:
            return
        self._pending.append(param.grad)
        self._pending_bytes += param.grad.numel() * param.grad.element_size()
        if self._pending_bytes >= self.bucket_bytes:
            self._flush()

    def _flush(self) -> None:
        if not self._pending:
            return
        flat = torch._utils._flatten_dense_tensors(self._pending)
        flat.div_(self.state.world_size)
        work = dist.all_reduce(flat, group=self.group, async_op=True)
        self._works.append((work, flat, list(self._pending)))
        self._pending = []
        self._pending_bytes = 0

    def finalize(self) -> None:
        """Wait for outstanding allreduces and scatter results back."""
        if not self.state.is_distributed:
            return
        self._flush()
        for work, flat, grads in self._works:
            work.wait()
            for g, synced in zip(
                grads, torch._utils._unflatten_dense_tensors(flat, grads)
            ):
                g.copy_(synced)
        self._works = []

    def remove(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks = []


def wrap_ddp(module: torch.nn.Module, process_group=None, bucket_bytes: int = 64 << 20):
    """Attach a GradBucketer to the module (stored as ``module._grad_bucketer``)."""
    state = DistributedState.get()
    if not state.is_distributed:
        return module
    if getattr(module, "_grad_bucketer", None) is None:
        module._grad_bucketer = GradBucketer(module, process_group, bucket_bytes)
    return module


def allreduce_gradients(module: torch.nn.Module) -> None:
    """Synchronous fused gradient allreduce (for modules without hooks)."""
    state = DistributedState.get()
    if not state.is_distributed:
        return
    bucketer = getattr(module, "_grad_bucketer", None)
    if bucketer is not None:
        bucketer.finalize()
        return
    grads = [p.grad for p in module.parameters() if p.grad is not None]
    if not grads:
        return
    flat = torch._utils._flatten_dense_tensors(grads)
    flat.div_(state.world_size)
    dist.all_reduce(flat)
    for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
        g.copy_(synced)


def broadcast_module(module: torch.nn.Module, src: int = 0, group=None) -> None:
    """Broadcast a module's parameters+buffers from ``src`` (flat, one op)."""
    state = DistributedState.get()
    if not state.is_distributed:
        return
    tensors = [t.data for t in module.parameters()] + list(module.buffers())
    tensors = [t for t in tensors if t.numel() > 0 and t.is_floating_point()]
    if not tensors:
        return
    flat = torch._utils._flatten_dense_tensors(tensors)
    dist.broadcast(flat, src=src, group=group)
    for t, synced in zip(tensors, torch._utils._unflatten_dense_tensors(flat, tensors)):
        t.copy_(synced)
