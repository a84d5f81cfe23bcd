"""Activation offload to pinned host memory.

Reference parity: ``agilerl/algorithms/core/base.py:4678``
(``_activation_offload_ctx`` via ``torch.autograd.graph.save_on_cpu``).
On MI355X the 288 GB of HBM makes this default-off, but
extreme-context runs can still opt in; pinned pages keep the
``hipMemcpyAsync`` H2D/D2H transfers off the compute stream's critical
path.
"""

from __future__ import annotations

from contextlib import contextmanager

import torch

__all__ = ["activation_offload"]


@contextmanager
def activation_offload(enabled: bool = True, pin_memory: bool = True):
    """Context manager: saved-for-backward activations live in pinned host
    RAM and stream back during backward."""
    if not enabled or not torch.cuda.is_available():
        yield
        return
    with torch.autograd.graph.save_on_cpu(pin_memory=pin_memory):
        yield
