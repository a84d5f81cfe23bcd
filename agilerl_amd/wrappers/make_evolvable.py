"""MakeEvolvable: adapt an arbitrary ``nn.Module`` into the evolvable API.

Reference parity: ``agilerl/wrappers/make_evolvable.py:42`` (deprecated in
the reference; kept for API compatibility).  A hooked forward pass
records the wrapped network's layer execution order and shapes
(the reference's ``detect_architecture``), then rebuilds it as the
matching evolvable module:

- Linear stacks        -> :class:`EvolvableMLP`   (layer/node mutations)
- Conv2d(+Linear) nets -> :class:`EvolvableCNN`   (layer/channel/kernel)
- LSTM(+Linear) nets   -> :class:`EvolvableLSTM`
- anything else        -> :class:`EvolvableWrapper` (no arch mutations)

Weights are seeded from the original layers wherever shapes line up, so
the evolvable copy starts from the wrapped network's function.
"""

from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn

from ..modules.base import EvolvableWrapper
from ..modules.cnn import EvolvableCNN
from ..modules.lstm import EvolvableLSTM
from ..modules.mlp import EvolvableMLP

__all__ = ["MakeEvolvable"]

_ACTIVATIONS = {
    nn.ReLU: "ReLU", nn.Tanh: "Tanh", nn.ELU: "ELU", nn.GELU: "GELU",
    nn.LeakyReLU: "LeakyReLU", nn.SiLU: "SiLU", nn.Sigmoid: "Sigmoid",
}


def _trace_layers(network: nn.Module, input_tensor: torch.Tensor):
    """Run one forward pass with hooks; return leaf layers in CALL order
    with their input/output shapes."""
    record: List[Tuple[nn.Module, tuple, tuple]] = []
    handles = []

    def hook(mod, inp, out):
        ishape = tuple(inp[0].shape) if inp and torch.is_tensor(inp[0]) else ()
        oshape = tuple(out.shape) if torch.is_tensor(out) else ()
        record.append((mod, ishape, oshape))

    for m in network.modules():
        if len(list(m.children())) == 0:  # leaf
            handles.append(m.register_forward_hook(hook))
    try:
        with torch.no_grad():
            network(input_tensor)
    finally:
        for h in handles:
            h.remove()
    return record


def _detect_activation(record) -> str:
    for mod, _, _ in record:
        for cls, name in _ACTIVATIONS.items():
            if type(mod) is cls:
                return name
    return "ReLU"


def _seed_matching(src_layers, dst_layers) -> None:
    with torch.no_grad():
        for src, dst in zip(src_layers, dst_layers):
            if src.weight.shape == dst.weight.shape:
                dst.weight.copy_(src.weight)
                if getattr(src, "bias", None) is not None and getattr(dst, "bias", None) is not None:
                    dst.bias.copy_(src.bias)


def MakeEvolvable(
    network: nn.Module,
    input_tensor: torch.Tensor,
    device: str = "cpu",
):
    """Returns an evolvable equivalent of ``network`` (see module doc)."""
    record = _trace_layers(network, input_tensor)
    ordered = [m for m, _, _ in record]
    linears = [m for m in ordered if isinstance(m, nn.Linear)]
    convs = [m for m in ordered if isinstance(m, nn.Conv2d)]
    lstms = [m for m in ordered if isinstance(m, nn.LSTM)]
    other_conv = any(isinstance(m, (nn.Conv1d, nn.Conv3d)) for m in ordered)
    activation = _detect_activation(record)

    with torch.no_grad():
        out = network(input_tensor)
    num_outputs = int(out.shape[-1])

    if convs and not lstms and not other_conv:
        # conv stack (+ linear head) -> EvolvableCNN
        input_shape = tuple(input_tensor.shape[1:])  # (C, H, W)
        evo = EvolvableCNN(
            input_shape=input_shape,
            num_outputs=num_outputs,
            channel_size=[c.out_channels for c in convs],
            kernel_size=[
                c.kernel_size[0] if isinstance(c.kernel_size, tuple) else c.kernel_size
                for c in convs
            ],
            stride_size=[
                c.stride[0] if isinstance(c.stride, tuple) else c.stride
                for c in convs
            ],
            activation=activation,
            device=device,
        )
        _seed_matching(convs, [m for m in evo.modules() if isinstance(m, nn.Conv2d)])
        _seed_matching(linears, [m for m in evo.modules() if isinstance(m, nn.Linear)])
        return evo

    if lstms and len(lstms) == 1 and not convs and not other_conv:
        lstm = lstms[0]
        evo = EvolvableLSTM(
            input_size=lstm.input_size,
            num_outputs=num_outputs,
            hidden_state_size=lstm.hidden_size,
            num_layers=lstm.num_layers,
            device=device,
        )
        with torch.no_grad():  # seed LSTM weights where shapes match
            src_sd = lstm.state_dict()
            dst = evo.lstm if hasattr(evo, "lstm") else None
            if dst is not None:
                dst_sd = dst.state_dict()
                for k in dst_sd:
                    if k in src_sd and src_sd[k].shape == dst_sd[k].shape:
                        dst_sd[k].copy_(src_sd[k])
        return evo

    if linears and not convs and not lstms and not other_conv:
        num_inputs = (
            int(input_tensor.reshape(input_tensor.shape[0], -1).shape[1])
            if input_tensor.dim() > 1
            else int(input_tensor.numel())
        )
        hidden = [l.out_features for l in linears[:-1]]
        evo = EvolvableMLP(
            num_inputs=num_inputs,
            num_outputs=num_outputs,
            hidden_size=hidden or [max(num_outputs, 16)],
            activation=activation,
            device=device,
        )
        _seed_matching(linears, [m for m in evo.modules() if isinstance(m, nn.Linear)])
        return evo

    return EvolvableWrapper(network, device=device)
