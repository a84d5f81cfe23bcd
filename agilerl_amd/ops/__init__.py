"""Hand-written CDNA4 kernel surface (with eager CPU references).

Each public function dispatches: GPU tensor -> HIP kernel (required, loud
failure if the extension is missing), CPU tensor -> eager PyTorch reference.

Kernel inventory (reference call sites in SURVEY.md §2.9):

- :func:`noisy_linear` — fused NoisyNet linear ``(mu+sigma*eps) @ x``.
- :func:`polyak_update_` — fused soft-update across a parameter list.
- :func:`gae_scan` — reverse-scan GAE advantages over (T, N).
- :func:`nstep_scan` — n-step return/bootstrapping for sampled windows.
- :func:`c51_project` — distributional-RL categorical projection.
- :func:`group_advantage` — GRPO group-relative advantage.
- :func:`masked_mean` — token-masked mean reduction.
- :func:`fused_linear_logprobs` — chunked lm_head logprob (fwd+bwd).
- :func:`grpo_token_loss` — fused token-masked GRPO/CISPO surrogate.

PER's GPU segment trees dispatch from
:mod:`agilerl_amd.components.segment_tree` to the ``segtree_update`` /
``segtree_retrieve`` / fused ``per_sample`` kernels in ``csrc/rl_ops.hip``
(single-workgroup barriered propagation; LDS-staged prefix descent).
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch
import torch.nn.functional as F

from .backend import extension, has_extension, require_extension, use_hip

__all__ = [
    "has_extension",
    "noisy_linear",
    "polyak_update_",
    "gae_scan",
    "nstep_scan",
    "c51_project",
    "group_advantage",
    "masked_mean",
    "masked_sum",
]


# ---------------------------------------------------------------------------
# NoisyLinear
# ---------------------------------------------------------------------------

def noisy_linear(
    x: torch.Tensor,
    weight_mu: torch.Tensor,
    weight_sigma: torch.Tensor,
    weight_epsilon: torch.Tensor,
    bias_mu: torch.Tensor,
    bias_sigma: torch.Tensor,
    bias_epsilon: torch.Tensor,
) -> torch.Tensor:
    """Train-time NoisyNet linear: ``x @ (mu + sigma*eps)^T + (bmu + bsig*beps)``.

    On GPU the effective-weight materialization is fused into the GEMM
    epilogue-free form via a HIP kernel (avoids writing W_eff to HBM).
    """
    ext = extension()
    if use_hip(x) and ext is not None and hasattr(ext, "noisy_linear"):
        return _NoisyLinearFn.apply(
            x, weight_mu, weight_sigma, weight_epsilon, bias_mu, bias_sigma, bias_epsilon
        )
    weight = weight_mu + weight_sigma * weight_epsilon
    bias = bias_mu + bias_sigma * bias_epsilon
    return F.linear(x, weight, bias)


class _NoisyLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w_mu, w_sig, w_eps, b_mu, b_sig, b_eps):
        ext = require_extension()
        out, w_eff = ext.noisy_linear_fwd(x, w_mu, w_sig, w_eps, b_mu, b_sig, b_eps)
        ctx.save_for_backward(x, w_eff, w_eps, b_eps)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x, w_eff, w_eps, b_eps = ctx.saved_tensors
        grad_out = grad_out.contiguous()
        x2 = x.reshape(-1, x.shape[-1])
        g2 = grad_out.reshape(-1, grad_out.shape[-1])
        grad_x = (g2 @ w_eff).reshape(x.shape)
        grad_w_mu = g2.t() @ x2
        grad_w_sig = grad_w_mu * w_eps
        grad_b_mu = g2.sum(0)
        grad_b_sig = grad_b_mu * b_eps
        return grad_x, grad_w_mu, grad_w_sig, None, grad_b_mu, grad_b_sig, None


# ---------------------------------------------------------------------------
# Polyak soft update
# ---------------------------------------------------------------------------

@torch.no_grad()
def polyak_update_(
    target_params: Sequence[torch.Tensor],
    source_params: Sequence[torch.Tensor],
    tau: float,
) -> None:
    """``target <- (1-tau)*target + tau*source`` over a whole parameter set.

    GPU path: a single HIP kernel over a flattened pointer list (one launch
    for the whole network instead of one lerp per tensor).
    """
    target_params = list(target_params)
    source_params = list(source_params)
    if not target_params:
        return
    ext = extension()
    if use_hip(*target_params) and ext is not None and hasattr(ext, "polyak_"):
        ext.polyak_(list(target_params), list(source_params), float(tau))
        return
    torch._foreach_lerp_(target_params, source_params, tau)


# ---------------------------------------------------------------------------
# GAE reverse scan
# ---------------------------------------------------------------------------

def gae_scan(
    rewards: torch.Tensor,
    values: torch.Tensor,
    dones: torch.Tensor,
    last_value: torch.Tensor,
    gamma: float,
    gae_lambda: float,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Generalized advantage estimation over a (T, N) rollout.

    ``dones[t]`` marks episode termination *after* step ``t`` — the
    convention every collector in this package stores — so step ``t``'s
    value bootstrap through ``V(s_{t+1})`` AND its lambda carry from
    ``adv[t+1]`` are both cut by ``dones[t]``: a terminal step's advantage
    is exactly ``r_t - V(s_t)``.  Returns ``(advantages, returns)`` with
    ``returns = advantages + values``.

    GPU path: one HIP kernel, one wavefront-lane per env column, the T-loop
    runs in-register (rollout tensors stay resident in HBM; the reference
    does this scan on CPU numpy — ``rollout_buffer.py:472``).
    """
    ext = extension()
    if use_hip(rewards) and ext is not None and hasattr(ext, "gae_scan"):
        adv = ext.gae_scan(
            rewards.float().contiguous(),
            values.float().contiguous(),
            dones.float().contiguous(),
            last_value.float().contiguous(),
            float(gamma),
            float(gae_lambda),
        )
        return adv, adv + values.float()

    T = rewards.shape[0]
    rewards = rewards.float()
    values = values.float()
    not_done = 1.0 - dones.float()
    adv = torch.zeros_like(rewards)
    next_adv = torch.zeros_like(last_value.float())
    next_value = last_value.float()
    for t in range(T - 1, -1, -1):
        nd = not_done[t]
        delta = rewards[t] + gamma * next_value * nd - values[t]
        next_adv = delta + gamma * gae_lambda * nd * next_adv
        adv[t] = next_adv
        next_value = values[t]
    return adv, adv + values


# ---------------------------------------------------------------------------
# n-step returns over sampled windows
# ---------------------------------------------------------------------------

def nstep_scan(
    rewards: torch.Tensor,
    dones: torch.Tensor,
    gamma: float,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """n-step return from per-sample windows.

    ``rewards``/``dones``: (B, n) windows of consecutive transitions starting
    at each sampled index.  Returns ``(returns, effective_steps)`` where
    ``returns[b] = sum_k gamma^k r[b,k]`` truncated at the first done, and
    ``effective_steps[b]`` is the number of steps accumulated (for the
    bootstrap ``gamma^steps * Q(s_{t+steps})`` and next-obs indexing).
    """
    ext = extension()
    if use_hip(rewards) and ext is not None and hasattr(ext, "nstep_scan"):
        return ext.nstep_scan(
            rewards.float().contiguous(), dones.float().contiguous(), float(gamma)
        )
    B, n = rewards.shape
    rewards = rewards.float()
    dones = dones.float()
    # alive[b,k] = 1 while no done strictly before step k
    alive = torch.cumprod(
        torch.cat([torch.ones(B, 1, device=dones.device), 1.0 - dones[:, :-1]], dim=1), dim=1
    )
    discounts = gamma ** torch.arange(n, device=rewards.device, dtype=torch.float32)
    returns = (rewards * alive * discounts).sum(dim=1)
    effective = alive.sum(dim=1)
    return returns, effective


# ---------------------------------------------------------------------------
# C51 categorical projection
# ---------------------------------------------------------------------------

def c51_project(
    next_dist: torch.Tensor,
    rewards: torch.Tensor,
    dones: torch.Tensor,
    support: torch.Tensor,
    gamma: float,
    v_min: float,
    v_max: float,
) -> torch.Tensor:
    """Project ``r + gamma*z`` onto the fixed support (distributional Bellman).

    next_dist: (B, A) probabilities over atoms; rewards/dones: (B,) or (B,1);
    support: (A,).  Returns the projected target distribution (B, A).
    """
    ext = extension()
    if use_hip(next_dist) and ext is not None and hasattr(ext, "c51_project"):
        return ext.c51_project(
            next_dist.float().contiguous(),
            rewards.float().reshape(-1).contiguous(),
            dones.float().reshape(-1).contiguous(),
            support.float().contiguous(),
            float(gamma),
            float(v_min),
            float(v_max),
        )
    B, A = next_dist.shape
    device = next_dist.device
    rewards = rewards.float().reshape(B, 1)
    dones = dones.float().reshape(B, 1)
    delta_z = (v_max - v_min) / (A - 1)
    tz = (rewards + (1.0 - dones) * gamma * support.view(1, A)).clamp_(v_min, v_max)
    b = (tz - v_min) / delta_z
    low = b.floor().long()
    up = b.ceil().long()
    # resolve low == up (b integral) so mass is not dropped
    eq = (up == low)
    low_adj = torch.where(eq & (low > 0), low - 1, low)
    up_adj = torch.where(eq & (low == 0), up + 1, up)
    proj = torch.zeros_like(next_dist)
    offset = (torch.arange(B, device=device) * A).view(B, 1)
    proj.view(-1).index_add_(
        0, (low_adj + offset).view(-1), (next_dist * (up_adj.float() - b)).view(-1)
    )
    proj.view(-1).index_add_(
        0, (up_adj.clamp_(max=A - 1) + offset).view(-1), (next_dist * (b - low_adj.float())).view(-1)
    )
    return proj


# ---------------------------------------------------------------------------
# GRPO group-relative advantage
# ---------------------------------------------------------------------------

def group_advantage(
    rewards: torch.Tensor, group_size: int, scale: bool = True, eps: float = 1e-8
) -> torch.Tensor:
    """Center each reward by its group mean (optionally /std). rewards: (B,)."""
    ext = extension()
    if use_hip(rewards) and ext is not None and hasattr(ext, "group_advantage"):
        return ext.group_advantage(rewards.float().contiguous(), int(group_size), bool(scale), float(eps))
    g = rewards.float().view(-1, group_size)
    adv = g - g.mean(dim=1, keepdim=True)
    if scale:
        adv = adv / (g.std(dim=1, keepdim=True) + eps)
    return adv.view(-1)


# ---------------------------------------------------------------------------
# Masked reductions (SURVEY §2.9.4 helpers)
# ---------------------------------------------------------------------------

def masked_mean(x: torch.Tensor, mask: torch.Tensor, dim: Optional[int] = None, eps: float = 1e-8):
    mask = mask.to(x.dtype)
    if dim is None:
        return (x * mask).sum() / (mask.sum() + eps)
    return (x * mask).sum(dim=dim) / (mask.sum(dim=dim) + eps)


def masked_sum(x: torch.Tensor, mask: torch.Tensor, dim: Optional[int] = None):
    mask = mask.to(x.dtype)
    return (x * mask).sum() if dim is None else (x * mask).sum(dim=dim)

from .swiglu import swiglu  # noqa: F401
from .rmsnorm import rms_norm, HipRMSNorm  # noqa: F401
