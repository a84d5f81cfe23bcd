"""MakeEvolvable: adapt an arbitrary ``nn.Module`` into the evolvable API.

Reference parity: ``agilerl/wrappers/make_evolvable.py:42`` (deprecated in
the reference; kept for API compatibility).  A forward pass introspects
the module's Linear stack; the rebuilt network is an
:class:`EvolvableMLP` with matching layer widths, seeded with the
original weights where shapes line up.  Non-sequential architectures fall
back to :class:`EvolvableWrapper` (no architecture mutations).
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from ..modules.base import EvolvableWrapper
from ..modules.mlp import EvolvableMLP

__all__ = ["MakeEvolvable"]


def MakeEvolvable(
    network: nn.Module,
    input_tensor: torch.Tensor,
    device: str = "cpu",
):
    """Returns an evolvable equivalent of ``network``."""
    linears: List[nn.Linear] = [m for m in network.modules() if isinstance(m, nn.Linear)]
    conv = any(isinstance(m, (nn.Conv1d, nn.Conv2d, nn.Conv3d)) for m in network.modules())
    recurrent = any(isinstance(m, (nn.LSTM, nn.GRU, nn.RNN)) for m in network.modules())
    if not linears or conv or recurrent:
        return EvolvableWrapper(network, device=device)

    with torch.no_grad():
        out = network(input_tensor)
    num_inputs = int(input_tensor.reshape(input_tensor.shape[0], -1).shape[1]) if input_tensor.dim() > 1 else int(input_tensor.numel())
    num_outputs = int(out.shape[-1])
    hidden = [l.out_features for l in linears[:-1]]

    # activation detection (first non-linear module after a Linear)
    activation = "ReLU"
    mods = list(network.modules())
    for i, m in enumerate(mods):
        if isinstance(m, nn.Tanh):
            activation = "Tanh"
            break
        if isinstance(m, nn.ELU):
            activation = "ELU"
            break
        if isinstance(m, (nn.GELU,)):
            activation = "GELU"
            break

    evo = EvolvableMLP(
        num_inputs=num_inputs,
        num_outputs=num_outputs,
        hidden_size=hidden or [max(num_outputs, 16)],
        activation=activation,
        device=device,
    )
    # seed weights from the original layer stack where shapes match
    evo_linears = [m for m in evo.model if isinstance(m, nn.Linear)]
    with torch.no_grad():
        for src, dst in zip(linears, evo_linears):
            if src.weight.shape == dst.weight.shape:
                dst.weight.copy_(src.weight)
                if src.bias is not None and dst.bias is not None:
                    dst.bias.copy_(src.bias)
    return evo
