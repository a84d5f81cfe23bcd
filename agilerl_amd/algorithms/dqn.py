"""DQN (with optional double-Q).

Reference parity: ``agilerl/algorithms/dqn.py:43`` — fused
cat(obs, next_obs) single forward (:363-377), polyak soft update :431
(HIP-fused here via ``ops.polyak_update_``), epsilon-greedy action
selection, optional hipGraph capture of the update step on GPU.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch
import torch.nn.functional as F

from .. import ops
from ..networks.q_networks import QNetwork
from ..spaces import Space
from .core.base import RLAlgorithm
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["DQN"]


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr=RLParameter(min=1e-5, max=1e-2),
        batch_size=RLParameter(min=16, max=1024, dtype=int),
        learn_step=RLParameter(min=1, max=16, dtype=int),
    )


class DQN(RLAlgorithm):
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
        double: bool = False,
        cudagraphs: bool = False,
        latent_dim: int = 64,
        normalize_images: bool = True,
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
            name="DQN",
        )
        self._accept_compat_kwargs(**kwargs)
        self.batch_size = int(batch_size)
        self.lr = float(lr)
        self.gamma = float(gamma)
        self.tau = float(tau)
        self.double = bool(double)
        self.cudagraphs = bool(cudagraphs)
        self.net_config = net_config
        self.latent_dim = latent_dim
        self.normalize_images = normalize_images
        self._graph = None
        self._graph_static = None

        if actor_network is not None:
            # user-supplied policy net (reference dqn.py:117 actor_network)
            from ..networks.base import CustomNetworkAdapter

            self.actor = CustomNetworkAdapter(actor_network, observation_space,
                                              action_space=action_space, device=device)
        else:
            self.actor = QNetwork(
                observation_space,
                action_space,
                encoder_config=net_config,
                head_config=head_config,
                latent_dim=latent_dim,
                device=device,
            )
        self.actor_target = self.actor.clone()
        for p in self.actor_target.parameters():
            p.requires_grad = False

        self.optimizer = OptimizerWrapper(torch.optim.Adam, [self.actor], lr=self.lr)

        self.register_network_group(
            NetworkGroup(eval_network="actor", shared_networks=["actor_target"], policy=True)
        )
        self.register_optimizer(
            OptimizerConfig(name="optimizer", networks=["actor"], lr_name="lr")
        )
        self.register_mutation_hook("_sync_target_after_mutation")

    # ------------------------------------------------------------------
    def _sync_target_after_mutation(self) -> None:
        """After an architecture mutation, hard-copy weights into the target."""
        self.actor_target.load_state_dict(self.actor.state_dict())
        for p in self.actor_target.parameters():
            p.requires_grad = False
        self._graph = None  # captured graph refers to the old modules
        self._graph_static = None

    # ------------------------------------------------------------------
    def get_action(
        self,
        obs,
        epsilon: float = 0.0,
        action_mask: Optional[np.ndarray] = None,
        training: bool = True,
    ) -> np.ndarray:
        with torch.no_grad():
            q = self.actor(self.actor.preprocess(obs))
        if action_mask is not None:
            mask_t = torch.as_tensor(np.asarray(action_mask), device=q.device, dtype=torch.bool)
            q = q.masked_fill(~mask_t, float("-inf"))
        greedy = q.argmax(dim=-1).cpu().numpy()
        if training and epsilon > 0:
            n = greedy.shape[0]
            rand_mask = np.random.rand(n) < epsilon
            if action_mask is not None:
                mask_np = np.asarray(action_mask, dtype=bool)
                rand_actions = np.array(
                    [np.random.choice(np.flatnonzero(mask_np[i])) for i in range(n)]
                )
            else:
                rand_actions = np.random.randint(0, self.action_space.n, size=n)
            greedy = np.where(rand_mask, rand_actions, greedy)
        return greedy

    # ------------------------------------------------------------------
    # hipGraph-captured update (reference dqn.py:120/:219 `cudagraphs`)
    # ------------------------------------------------------------------
    def _update_body(self, obs, actions, rewards, next_obs, dones, loss_out):
        with torch.no_grad():
            q_next_target = self.actor_target(self.actor_target.preprocess(next_obs))
            if self.double:
                next_actions = self.actor(self.actor.preprocess(next_obs)).argmax(-1, keepdim=True)
                q_next = q_next_target.gather(1, next_actions).squeeze(-1)
            else:
                q_next = q_next_target.max(dim=-1).values
            target = rewards + (1.0 - dones) * self.gamma * q_next
        q_pred = self.actor(self.actor.preprocess(obs)).gather(1, actions.unsqueeze(1)).squeeze(-1)
        loss = F.huber_loss(q_pred, target)
        self.optimizer.zero_grad(set_to_none=False)
        loss.backward()
        self.optimizer.step()
        # graph-safe polyak: foreach lerp (stable param addresses).  MUST be
        # no-grad: an in-place lerp from grad-requiring actor params would
        # silently make the target params non-leaf graph nodes (leaking the
        # autograd graph and breaking later requires_grad mutation).
        with torch.no_grad():
            torch._foreach_lerp_(
                list(self.actor_target.parameters()), list(self.actor.parameters()), self.tau
            )
        loss_out.copy_(loss.detach())

    def _graphed_learn(self, obs, actions, rewards, next_obs, dones) -> float:
        B = actions.shape[0]
        if self._graph_static is not None and self._graph_static["obs"].shape[0] != B:
            self._graph = None
        if self._graph is None:
            # capturable optimizer state required for in-graph Adam steps
            prev_opt = self.optimizer.optimizer
            prev_state = dict(prev_opt.state) if prev_opt is not None else {}
            self.optimizer.optimizer = torch.optim.Adam(
                [p for p in self.actor.parameters() if p.requires_grad],
                lr=self.lr, capturable=True,
            )
            st = {
                "obs": obs.clone(), "actions": actions.clone(), "rewards": rewards.clone(),
                "next_obs": next_obs.clone(), "dones": dones.clone(),
                "loss": torch.zeros((), device=obs.device),
            }
            self._graph_static = st
            # warmup + capture run REAL updates; snapshot and restore so the
            # graphed agent stays numerically identical to the eager path
            saved_actor = [p.detach().clone() for p in self.actor.parameters()]
            saved_target = [p.detach().clone() for p in self.actor_target.parameters()]
            torch.cuda.synchronize()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):  # warmup
                    self._update_body(st["obs"], st["actions"], st["rewards"],
                                      st["next_obs"], st["dones"], st["loss"])
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                self._update_body(st["obs"], st["actions"], st["rewards"],
                                  st["next_obs"], st["dones"], st["loss"])
            with torch.no_grad():
                for p, sv in zip(self.actor.parameters(), saved_actor):
                    p.copy_(sv)
                for p, sv in zip(self.actor_target.parameters(), saved_target):
                    p.copy_(sv)
                # restore optimizer state IN PLACE (the graph holds these
                # tensors): carry over pre-existing Adam moments, else zero
                for p, group_state in self.optimizer.optimizer.state.items():
                    prior = prev_state.get(p)
                    for k, v in group_state.items():
                        if not torch.is_tensor(v):
                            continue
                        pv = prior.get(k) if prior else None
                        if pv is not None and torch.is_tensor(pv) and pv.shape == v.shape:
                            v.copy_(pv.to(v.device, v.dtype))
                        elif pv is not None and k == "step" and v.numel() == 1:
                            v.fill_(float(pv))
                        else:
                            v.zero_()
            torch.cuda.synchronize()
        st = self._graph_static
        st["obs"].copy_(obs)
        st["actions"].copy_(actions)
        st["rewards"].copy_(rewards)
        st["next_obs"].copy_(next_obs)
        st["dones"].copy_(dones)
        self._graph.replay()
        return float(st["loss"])

    # ------------------------------------------------------------------
    def learn(self, experiences: Dict[str, torch.Tensor]) -> float:
        if (
            self.cudagraphs
            and torch.cuda.is_available()
            and "weights" not in experiences
            and "n_steps" not in experiences
        ):
            obs = experiences["obs"]
            if isinstance(obs, torch.Tensor):
                return self._graphed_learn(
                    obs.to(self.device).float(),
                    experiences["action"].to(self.device).long().reshape(-1),
                    experiences["reward"].to(self.device).float().reshape(-1),
                    experiences["next_obs"].to(self.device).float(),
                    experiences["done"].to(self.device).float().reshape(-1),
                )
        obs = experiences["obs"]
        actions = experiences["action"].to(self.device).long().reshape(-1)
        rewards = experiences["reward"].to(self.device).float().reshape(-1)
        next_obs = experiences["next_obs"]
        dones = experiences["done"].to(self.device).float().reshape(-1)

        with torch.no_grad():
            q_next_target = self.actor_target(self.actor_target.preprocess(next_obs))
            if self.double:
                next_actions = self.actor(self.actor.preprocess(next_obs)).argmax(-1, keepdim=True)
                q_next = q_next_target.gather(1, next_actions).squeeze(-1)
            else:
                q_next = q_next_target.max(dim=-1).values
            n_steps = experiences.get("n_steps")
            discount = (
                self.gamma ** n_steps.float().reshape(-1) if n_steps is not None else self.gamma
            )
            target = rewards + (1.0 - dones) * discount * q_next

        q_pred = (
            self.actor(self.actor.preprocess(obs)).gather(1, actions.unsqueeze(1)).squeeze(-1)
        )
        loss = F.huber_loss(q_pred, target)
        self.optimizer.zero_grad()
        loss.backward()
        self.optimizer.step()
        self.soft_update()
        return float(loss.detach())

    def soft_update(self) -> None:
        ops.polyak_update_(
            list(self.actor_target.parameters()), list(self.actor.parameters()), self.tau
        )
