"""Rainbow DQN: double + dueling + distributional (C51) + noisy + n-step + PER.

Reference parity: ``agilerl/algorithms/dqn_rainbow.py:42`` — C51 target
projection (:389-396, here the HIP ``ops.c51_project`` kernel), NoisyLinear
exploration (fused noisy-GEMM on GPU), per-sample loss -> PER priorities.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from .. import ops
from ..networks.q_networks import RainbowQNetwork
from ..spaces import Space
from .core.base import RLAlgorithm
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["RainbowDQN"]


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr=RLParameter(min=1e-5, max=1e-2),
        batch_size=RLParameter(min=16, max=1024, dtype=int),
        learn_step=RLParameter(min=1, max=16, dtype=int),
    )


class RainbowDQN(RLAlgorithm):
    def __init__(
        self,
        observation_space: Space,
        action_space: Space,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        net_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        batch_size: int = 64,
        lr: float = 1e-4,
        learn_step: int = 5,
        gamma: float = 0.99,
        tau: float = 1e-3,
        beta: float = 0.4,
        prior_eps: float = 1e-6,
        num_atoms: int = 51,
        v_min: float = -10.0,
        v_max: float = 10.0,
        n_step: int = 3,
        noise_std: float = 0.5,
        combined_reward: bool = False,
        latent_dim: int = 64,
        actor_network=None,
        device: str = "cpu",
        **kwargs,
    ):
        super().__init__(
            observation_space,
            action_space,
            index=index,
            learn_step=learn_step,
            device=device,
            hp_config=hp_config or default_hp_config(),
            name="RainbowDQN",
        )
        self._accept_compat_kwargs(**kwargs)
        self.batch_size = int(batch_size)
        self.lr = float(lr)
        self.gamma = float(gamma)
        self.tau = float(tau)
        self.beta = float(beta)
        self.prior_eps = float(prior_eps)
        self.num_atoms = int(num_atoms)
        self.v_min = float(v_min)
        self.v_max = float(v_max)
        self.n_step = int(n_step)
        self.noise_std = noise_std
        # reference dqn_rainbow.py:124: sum the 1-step and n-step losses
        # (requires the buffer to emit the *_1step fields; the off-policy
        # loop passes include_one_step when this flag is set)
        self.combined_reward = bool(combined_reward)
        self.net_config = net_config
        self.latent_dim = latent_dim
        self.last_td_errors: Optional[torch.Tensor] = None

        if actor_network is not None:
            # user-supplied distributional Q net: must expose .dist()/.support
            # like RainbowQNetwork (reference dqn_rainbow.py:125)
            if not (hasattr(actor_network, "dist") and hasattr(actor_network, "support")):
                raise TypeError(
                    "RainbowDQN actor_network must expose .dist(obs) and "
                    ".support (see networks.RainbowQNetwork)"
                )
            self.actor = actor_network.to(device)
        else:
            self.actor = RainbowQNetwork(
                observation_space,
                action_space,
                encoder_config=net_config,
                head_config=head_config,
                latent_dim=latent_dim,
                num_atoms=num_atoms,
                v_min=v_min,
                v_max=v_max,
                noise_std=noise_std,
                device=device,
            )
        self.actor_target = self.actor.clone()
        for p in self.actor_target.parameters():
            p.requires_grad = False

        self.optimizer = OptimizerWrapper(torch.optim.Adam, [self.actor], lr=self.lr)

        self.register_network_group(
            NetworkGroup(eval_network="actor", shared_networks=["actor_target"], policy=True)
        )
        self.register_optimizer(OptimizerConfig(name="optimizer", networks=["actor"], lr_name="lr"))
        self.register_mutation_hook("_sync_target_after_mutation")

    def _sync_target_after_mutation(self) -> None:
        self.actor_target.load_state_dict(self.actor.state_dict())
        for p in self.actor_target.parameters():
            p.requires_grad = False

    # ------------------------------------------------------------------
    def get_action(
        self,
        obs,
        action_mask: Optional[np.ndarray] = None,
        training: bool = True,
        **kwargs,
    ) -> np.ndarray:
        """Noisy-network exploration: no epsilon; fresh noise each call."""
        if training:
            self.actor.reset_noise()
            self.actor.train()
        else:
            self.actor.eval()
        with torch.no_grad():
            q = self.actor(self.actor.preprocess(obs))
        if action_mask is not None:
            mask_t = torch.as_tensor(np.asarray(action_mask), device=q.device, dtype=torch.bool)
            q = q.masked_fill(~mask_t, float("-inf"))
        greedy = q.argmax(dim=-1)
        if isinstance(obs, torch.Tensor) and obs.is_cuda:
            return greedy  # device-native path (torch envs)
        return greedy.cpu().numpy()

    # ------------------------------------------------------------------
    def learn(self, experiences: Dict[str, torch.Tensor]) -> float:
        obs = experiences["obs"]
        actions = experiences["action"].to(self.device).long().reshape(-1)
        rewards = experiences["reward"].to(self.device).float().reshape(-1)
        next_obs = experiences["next_obs"]
        dones = experiences["done"].to(self.device).float().reshape(-1)
        weights = experiences.get("weights")
        n_steps = experiences.get("n_steps")

        B = actions.shape[0]
        self.actor.train()
        self.actor_target.train()

        with torch.no_grad():
            self.actor.reset_noise()
            self.actor_target.reset_noise()
            # double-DQN action selection on the online net
            next_q = self.actor(self.actor.preprocess(next_obs))
            next_actions = next_q.argmax(dim=-1)
            next_dist = self.actor_target.dist(self.actor_target.preprocess(next_obs))
            next_dist = next_dist[torch.arange(B, device=next_dist.device), next_actions]
            gamma_n = (
                self.gamma ** n_steps.float().reshape(-1)
                if n_steps is not None
                else torch.full_like(rewards, self.gamma)
            )
            # projection with per-sample effective discount: fold gamma^n by
            # scaling the support through the kernel per sample is equivalent
            # to projecting r + gamma_n * z
            target_dist = self._project(next_dist, rewards, dones, gamma_n)

        dist = self.actor.dist(self.actor.preprocess(obs))
        log_p = torch.log(dist[torch.arange(B, device=dist.device), actions])
        elementwise_loss = -(target_dist * log_p).sum(-1)

        if self.combined_reward and "reward_1step" in experiences:
            # reference dqn_rainbow.py:453: add the 1-step loss for the same
            # anchors to the n-step loss (1-step fields come from the buffer's
            # include_one_step sampling mode)
            r1 = experiences["reward_1step"].to(self.device).float().reshape(-1)
            no1 = experiences["next_obs_1step"]
            d1 = experiences["done_1step"].to(self.device).float().reshape(-1)
            with torch.no_grad():
                nq1 = self.actor(self.actor.preprocess(no1))
                na1 = nq1.argmax(dim=-1)
                nd1 = self.actor_target.dist(self.actor_target.preprocess(no1))
                nd1 = nd1[torch.arange(B, device=nd1.device), na1]
                target_1 = self._project(nd1, r1, d1, torch.full_like(r1, self.gamma))
            elementwise_loss = elementwise_loss - (target_1 * log_p).sum(-1)

        if weights is not None:
            loss = (elementwise_loss * weights.to(self.device).reshape(-1)).mean()
        else:
            loss = elementwise_loss.mean()

        self.optimizer.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.actor.parameters(), 10.0)
        self.optimizer.step()

        self.last_td_errors = elementwise_loss.detach() + self.prior_eps
        ops.polyak_update_(
            list(self.actor_target.parameters()), list(self.actor.parameters()), self.tau
        )
        return float(loss.detach())

    def _project(
        self,
        next_dist: torch.Tensor,
        rewards: torch.Tensor,
        dones: torch.Tensor,
        gamma_n: torch.Tensor,
    ) -> torch.Tensor:
        """C51 projection; uniform gamma uses the fused kernel, per-sample
        gamma (mixed n-step windows) falls back to the batched eager path."""
        support = self.actor.support
        if torch.all(gamma_n == gamma_n[0]):
            return ops.c51_project(
                next_dist, rewards, dones, support, float(gamma_n[0]), self.v_min, self.v_max
            )
        B, A = next_dist.shape
        delta_z = (self.v_max - self.v_min) / (A - 1)
        tz = (rewards.view(B, 1) + (1 - dones.view(B, 1)) * gamma_n.view(B, 1) * support.view(1, A)).clamp(
            self.v_min, self.v_max
        )
        b = (tz - self.v_min) / delta_z
        low = b.floor().long()
        up = b.ceil().long()
        eq = up == low
        low_adj = torch.where(eq & (low > 0), low - 1, low)
        up_adj = torch.where(eq & (low == 0), up + 1, up)
        proj = torch.zeros_like(next_dist)
        offset = (torch.arange(B, device=next_dist.device) * A).view(B, 1)
        proj.view(-1).index_add_(0, (low_adj + offset).view(-1), (next_dist * (up_adj.float() - b)).view(-1))
        proj.view(-1).index_add_(
            0, (up_adj.clamp_(max=A - 1) + offset).view(-1), (next_dist * (b - low_adj.float())).view(-1)
        )
        return proj
