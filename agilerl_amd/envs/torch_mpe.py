"""Device-resident multi-agent particle envs (torch ports of envs/mpe.py).

Same scenarios/dynamics as the numpy MPE envs but batched in HBM — for
the MADDPG/MATD3 GPU path the dict observations, rewards and resets stay
on-device (the CPU-env profile was dominated by per-step host work).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from ..spaces import Box, Discrete, Space

__all__ = ["SpeakerListenerTorchVecEnv", "SimpleSpreadTorchVecEnv"]

DT = 0.1
DAMPING = 0.25
ACCEL = 5.0
MAX_SPEED = 1.3


class _TorchMABase:
    is_torch = True
    max_episode_steps = 25

    agents: List[str]
    observation_spaces: Dict[str, Space]
    action_spaces: Dict[str, Space]

    def __init__(self, num_envs: int, device: str = "cuda", seed: Optional[int] = None):
        self.num_envs = int(num_envs)
        self.device = device
        self.gen = torch.Generator(device=device)
        if seed is not None:
            self.gen.manual_seed(seed)
        self._elapsed = torch.zeros(self.num_envs, dtype=torch.long, device=device)

    def _rand(self, *shape, low=-1.0, high=1.0) -> torch.Tensor:
        u = torch.rand(*shape, generator=self.gen, device=self.device)
        return low + u * (high - low)

    @property
    def possible_agents(self):
        return self.agents

    def observation_space(self, agent):
        return self.observation_spaces[agent]

    def action_space(self, agent):
        return self.action_spaces[agent]

    @staticmethod
    def _move(vel: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
        vel = vel * (1.0 - DAMPING) + u * ACCEL * DT
        speed = vel.norm(dim=-1, keepdim=True)
        return vel * torch.where(speed > MAX_SPEED, MAX_SPEED / speed.clamp(min=1e-8),
                                 torch.ones_like(speed))

    @staticmethod
    def _discrete_to_force(a: torch.Tensor) -> torch.Tensor:
        fx = torch.where(a == 1, -1.0, torch.where(a == 2, 1.0, torch.zeros_like(a, dtype=torch.float32)))
        fy = torch.where(a == 3, -1.0, torch.where(a == 4, 1.0, torch.zeros_like(a, dtype=torch.float32)))
        return torch.stack([fx, fy], dim=-1)

    # hooks
    def _reset_rows(self, mask: torch.Tensor) -> None:
        raise NotImplementedError

    def _obs(self) -> Dict[str, torch.Tensor]:
        raise NotImplementedError

    def _step_all(self, actions) -> Dict[str, torch.Tensor]:
        raise NotImplementedError

    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self.gen.manual_seed(seed)
        self._reset_rows(torch.ones(self.num_envs, dtype=torch.bool, device=self.device))
        self._elapsed.zero_()
        return self._obs(), {}

    def step(self, actions: Dict[str, torch.Tensor]):
        rewards = self._step_all(actions)
        self._elapsed += 1
        trunc_arr = self._elapsed >= self.max_episode_steps
        term = {a: torch.zeros(self.num_envs, dtype=torch.bool, device=self.device)
                for a in self.agents}
        trunc = {a: trunc_arr.clone() for a in self.agents}
        obs = self._obs()
        info: Dict = {"final_observation": obs, "done_mask": trunc_arr}
        self._reset_rows(trunc_arr)
        self._elapsed.copy_(torch.where(trunc_arr, torch.zeros_like(self._elapsed), self._elapsed))
        fresh = self._obs()
        m = trunc_arr.view(-1, 1)
        obs = {a: torch.where(m, fresh[a], obs[a]) for a in self.agents}
        return obs, rewards, term, trunc, info


class SpeakerListenerTorchVecEnv(_TorchMABase):
    N_LANDMARKS = 3

    def __init__(self, num_envs: int = 1, device: str = "cuda",
                 seed: Optional[int] = None, continuous_actions: bool = False):
        super().__init__(num_envs, device, seed)
        self.continuous_actions = continuous_actions
        self.agents = ["speaker_0", "listener_0"]
        self.observation_spaces = {
            "speaker_0": Box(-np.inf, np.inf, (3,)),
            "listener_0": Box(-np.inf, np.inf, (11,)),
        }
        self.action_spaces = {"speaker_0": Discrete(3), "listener_0": Discrete(5)}
        N = self.num_envs
        self.listener_pos = torch.zeros(N, 2, device=device)
        self.listener_vel = torch.zeros(N, 2, device=device)
        self.landmarks = torch.zeros(N, self.N_LANDMARKS, 2, device=device)
        self.goal = torch.zeros(N, dtype=torch.long, device=device)
        self.comm = torch.zeros(N, 3, device=device)

    def _reset_rows(self, mask: torch.Tensor) -> None:
        N = self.num_envs
        m1, m2, m3 = mask.view(-1, 1), mask.view(-1, 1, 1), mask
        self.listener_pos.copy_(torch.where(m1, self._rand(N, 2), self.listener_pos))
        self.listener_vel.copy_(torch.where(m1, torch.zeros_like(self.listener_vel), self.listener_vel))
        self.landmarks.copy_(torch.where(m2, self._rand(N, self.N_LANDMARKS, 2), self.landmarks))
        new_goal = (self._rand(N, low=0.0, high=float(self.N_LANDMARKS)).long()
                    .clamp(max=self.N_LANDMARKS - 1))
        self.goal.copy_(torch.where(m3, new_goal, self.goal))
        self.comm.copy_(torch.where(m1, torch.zeros_like(self.comm), self.comm))

    def _obs(self) -> Dict[str, torch.Tensor]:
        N = self.num_envs
        goal_onehot = torch.nn.functional.one_hot(self.goal, self.N_LANDMARKS).float()
        rel = (self.landmarks - self.listener_pos.unsqueeze(1)).reshape(N, -1)
        return {
            "speaker_0": goal_onehot,
            "listener_0": torch.cat([self.listener_vel, rel, self.comm], dim=1),
        }

    def _step_all(self, actions):
        sp = actions["speaker_0"].reshape(-1).long()
        li = actions["listener_0"].reshape(-1).long()
        self.comm.copy_(torch.nn.functional.one_hot(sp.clamp(0, 2), 3).float())
        u = self._discrete_to_force(li)
        self.listener_vel.copy_(self._move(self.listener_vel, u))
        self.listener_pos.add_(self.listener_vel * DT)
        goal_pos = self.landmarks.gather(
            1, self.goal.view(-1, 1, 1).expand(-1, 1, 2)
        ).squeeze(1)
        dist = (self.listener_pos - goal_pos).norm(dim=1)
        reward = -dist
        return {a: reward.clone() for a in self.agents}


class SimpleSpreadTorchVecEnv(_TorchMABase):
    N_AGENTS = 3
    N_LANDMARKS = 3
    AGENT_SIZE = 0.15

    def __init__(self, num_envs: int = 1, device: str = "cuda",
                 seed: Optional[int] = None, continuous_actions: bool = False):
        super().__init__(num_envs, device, seed)
        self.agents = [f"agent_{i}" for i in range(self.N_AGENTS)]
        obs_dim = 4 + 2 * self.N_LANDMARKS + 2 * (self.N_AGENTS - 1)
        self.observation_spaces = {a: Box(-np.inf, np.inf, (obs_dim,)) for a in self.agents}
        self.action_spaces = {a: Discrete(5) for a in self.agents}
        N = self.num_envs
        self.pos = torch.zeros(N, self.N_AGENTS, 2, device=device)
        self.vel = torch.zeros(N, self.N_AGENTS, 2, device=device)
        self.landmarks = torch.zeros(N, self.N_LANDMARKS, 2, device=device)

    def _reset_rows(self, mask: torch.Tensor) -> None:
        N = self.num_envs
        m = mask.view(-1, 1, 1)
        self.pos.copy_(torch.where(m, self._rand(N, self.N_AGENTS, 2), self.pos))
        self.vel.copy_(torch.where(m, torch.zeros_like(self.vel), self.vel))
        self.landmarks.copy_(torch.where(m, self._rand(N, self.N_LANDMARKS, 2), self.landmarks))

    def _obs(self) -> Dict[str, torch.Tensor]:
        N = self.num_envs
        out = {}
        for i, name in enumerate(self.agents):
            rel_lm = (self.landmarks - self.pos[:, i : i + 1]).reshape(N, -1)
            others = [j for j in range(self.N_AGENTS) if j != i]
            rel_ag = (self.pos[:, others] - self.pos[:, i : i + 1]).reshape(N, -1)
            out[name] = torch.cat([self.vel[:, i], self.pos[:, i], rel_lm, rel_ag], dim=1)
        return out

    def _step_all(self, actions):
        for i, name in enumerate(self.agents):
            u = self._discrete_to_force(actions[name].reshape(-1).long())
            self.vel[:, i] = self._move(self.vel[:, i], u)
            self.pos[:, i] = self.pos[:, i] + self.vel[:, i] * DT
        d = (self.pos.unsqueeze(2) - self.landmarks.unsqueeze(1)).norm(dim=-1)
        reward = -d.min(dim=1).values.sum(dim=1)
        pd = (self.pos.unsqueeze(2) - self.pos.unsqueeze(1)).norm(dim=-1)
        ii = torch.arange(self.N_AGENTS, device=self.device)
        pd[:, ii, ii] = float("inf")
        collisions = (pd < 2 * self.AGENT_SIZE).sum(dim=(1, 2)).float() / 2
        reward = reward - collisions
        return {a: reward.clone() for a in self.agents}
