"""Custom layers used by the evolvable architectures.

Reference parity: ``agilerl/modules/custom_components.py`` (NoisyLinear :41,
GumbelSoftmax :13, NewGELU :137, ResidualBlock :155, SimbaResidualBlock :227).

``NoisyLinear`` is a HIP-kernel target on MI355X (fused noisy-GEMM; see
``agilerl_amd/ops``). The module-level implementation here is the portable
reference path; the op layer dispatches to the CDNA4 kernel on gfx950.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = [
    "NoisyLinear",
    "GumbelSoftmax",
    "NewGELU",
    "ResidualBlock",
    "SimbaResidualBlock",
    "get_activation",
    "ACTIVATION_REGISTRY",
]


ACTIVATION_REGISTRY = {
    "ReLU": nn.ReLU,
    "Tanh": nn.Tanh,
    "Sigmoid": nn.Sigmoid,
    "ELU": nn.ELU,
    "LeakyReLU": nn.LeakyReLU,
    "GELU": nn.GELU,
    "SiLU": nn.SiLU,
    "Softsign": nn.Softsign,
    "Softplus": nn.Softplus,
    "PReLU": nn.PReLU,
    "Identity": nn.Identity,
    "Mish": nn.Mish,
}


def get_activation(name: Optional[str]) -> nn.Module:
    if name is None or name == "None":
        return nn.Identity()
    if name == "NewGELU":
        return NewGELU()
    if name == "GumbelSoftmax":
        return GumbelSoftmax()
    try:
        return ACTIVATION_REGISTRY[name]()
    except KeyError as e:
        raise ValueError(f"Unknown activation '{name}'. Options: {sorted(ACTIVATION_REGISTRY)}") from e


class NoisyLinear(nn.Module):
    """Linear layer with factorized Gaussian parameter noise (NoisyNets).

    Train-time weights are ``mu + sigma * eps`` where ``eps`` is the outer
    product of two factorized noise vectors passed through
    ``f(x) = sign(x) * sqrt(|x|)``.
    """

    def __init__(self, in_features: int, out_features: int, std_init: float = 0.5,
                 device=None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.std_init = std_init

        self.weight_mu = nn.Parameter(torch.empty(out_features, in_features))
        self.weight_sigma = nn.Parameter(torch.empty(out_features, in_features))
        self.register_buffer("weight_epsilon", torch.zeros(out_features, in_features))
        self.bias_mu = nn.Parameter(torch.empty(out_features))
        self.bias_sigma = nn.Parameter(torch.empty(out_features))
        self.register_buffer("bias_epsilon", torch.zeros(out_features))

        self.reset_parameters()
        self.reset_noise()
        if device is not None:
            self.to(device)

    def reset_parameters(self) -> None:
        bound = 1.0 / math.sqrt(self.in_features)
        self.weight_mu.data.uniform_(-bound, bound)
        self.weight_sigma.data.fill_(self.std_init / math.sqrt(self.in_features))
        self.bias_mu.data.uniform_(-bound, bound)
        self.bias_sigma.data.fill_(self.std_init / math.sqrt(self.out_features))

    @staticmethod
    def _scaled_noise(size: int, device, generator=None) -> torch.Tensor:
        x = torch.randn(size, device=device, generator=generator)
        return x.sign() * x.abs().sqrt()

    @torch.no_grad()
    def reset_noise(self, generator: Optional[torch.Generator] = None) -> None:
        eps_in = self._scaled_noise(self.in_features, self.weight_mu.device, generator)
        eps_out = self._scaled_noise(self.out_features, self.weight_mu.device, generator)
        self.weight_epsilon.copy_(torch.outer(eps_out, eps_in))
        self.bias_epsilon.copy_(eps_out)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.training:
            from ..ops import noisy_linear as _noisy_op

            return _noisy_op(
                x,
                self.weight_mu,
                self.weight_sigma,
                self.weight_epsilon,
                self.bias_mu,
                self.bias_sigma,
                self.bias_epsilon,
            )
        return F.linear(x, self.weight_mu, self.bias_mu)


class GumbelSoftmax(nn.Module):
    """Differentiable sample from a categorical via Gumbel-Softmax."""

    def __init__(self, tau: float = 1.0, hard: bool = True, eps: float = 1e-10):
        super().__init__()
        self.tau = tau
        self.hard = hard
        self.eps = eps

    def forward(self, logits: torch.Tensor) -> torch.Tensor:
        if self.training:
            return F.gumbel_softmax(logits, tau=self.tau, hard=self.hard, eps=self.eps, dim=-1)
        index = logits.argmax(dim=-1, keepdim=True)
        return torch.zeros_like(logits).scatter_(-1, index, 1.0)


class NewGELU(nn.Module):
    """GPT-2 style tanh-approximated GELU."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return (
            0.5
            * x
            * (1.0 + torch.tanh(math.sqrt(2.0 / math.pi) * (x + 0.044715 * torch.pow(x, 3.0))))
        )


class ResidualBlock(nn.Module):
    """Conv residual block (used by EvolvableResNet)."""

    def __init__(self, channels: int, kernel_size: int = 3, stride: int = 1):
        super().__init__()
        pad = kernel_size // 2
        self.conv1 = nn.Conv2d(channels, channels, kernel_size, stride, pad)
        self.bn1 = nn.BatchNorm2d(channels)
        self.conv2 = nn.Conv2d(channels, channels, kernel_size, stride, pad)
        self.bn2 = nn.BatchNorm2d(channels)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return F.relu(out + x)


class SimbaResidualBlock(nn.Module):
    """Pre-LayerNorm MLP residual block (SimBa architecture)."""

    def __init__(self, hidden_size: int, scale_factor: int = 4):
        super().__init__()
        self.ln = nn.LayerNorm(hidden_size)
        self.fc1 = nn.Linear(hidden_size, hidden_size * scale_factor)
        self.fc2 = nn.Linear(hidden_size * scale_factor, hidden_size)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = self.ln(x)
        h = F.relu(self.fc1(h))
        return x + self.fc2(h)
