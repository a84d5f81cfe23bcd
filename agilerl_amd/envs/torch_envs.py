"""Device-resident torch vector environments.

MI355X-native design: for GPU training the env itself lives in HBM —
observations, rewards and resets are torch ops on-device, so the
collect loop never crosses PCIe (the rocprof profile of the CPU-env
bench showed __amd_rocclr_copyBuffer as the top kernel: 66k tiny
per-step transfers; see profiles/r01_notes.md).  The API mirrors the
numpy ``BatchedVecEnv`` but returns torch tensors; ``is_torch = True``
signals the device-native collect path.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from ..spaces import Box, Discrete
import numpy as np

__all__ = ["TorchVecEnv", "LunarLanderTorchVecEnv", "CartPoleTorchVecEnv"]


class TorchVecEnv:
    is_torch = True
    max_episode_steps: Optional[int] = None

    def __init__(self, num_envs: int, device: str = "cuda", seed: Optional[int] = None):
        self.num_envs = int(num_envs)
        self.device = device
        self.gen = torch.Generator(device=device)
        if seed is not None:
            self.gen.manual_seed(seed)
        self._elapsed = torch.zeros(self.num_envs, dtype=torch.long, device=device)
        self._ep_return = torch.zeros(self.num_envs, device=device)

    # hooks --------------------------------------------------------------
    def _reset_rows(self, mask: torch.Tensor) -> None:
        raise NotImplementedError

    def _obs(self) -> torch.Tensor:
        raise NotImplementedError

    def _step_all(self, actions: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        raise NotImplementedError

    def _rand(self, *shape, low=0.0, high=1.0) -> torch.Tensor:
        # graph_safe: hipGraph capture requires the default (graph-registered)
        # CUDA generator rather than a user Generator object
        gen = None if getattr(self, "graph_safe", False) else self.gen
        u = torch.rand(*shape, generator=gen, device=self.device)
        return low + u * (high - low)

    # API ----------------------------------------------------------------
    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self.gen.manual_seed(seed)
        self._reset_rows(torch.ones(self.num_envs, dtype=torch.bool, device=self.device))
        self._elapsed.zero_()
        self._ep_return.zero_()
        return self._obs(), {}

    def step(self, actions: torch.Tensor):
        reward, terminated = self._step_all(actions)
        self._elapsed += 1
        if self.max_episode_steps is not None:
            truncated = (self._elapsed >= self.max_episode_steps) & ~terminated
        else:
            truncated = torch.zeros_like(terminated)
        done = terminated | truncated
        obs = self._obs()
        self._ep_return += reward
        info: Dict = {}
        # branchless auto-reset: masked reset every step (no host sync);
        # all state updates are IN-PLACE so the whole step is
        # hipGraph-capturable (stable tensor addresses across replays)
        info["final_observation"] = obs
        info["episode_return"] = self._ep_return.clone()
        info["done_mask"] = done
        self._reset_rows(done)
        self._elapsed.copy_(torch.where(done, torch.zeros_like(self._elapsed), self._elapsed))
        self._ep_return.copy_(torch.where(done, torch.zeros_like(self._ep_return), self._ep_return))
        done_b = done.view(-1, *([1] * (obs.dim() - 1)))
        obs = torch.where(done_b, self._obs(), obs)
        return obs, reward, terminated, truncated, info


class LunarLanderTorchVecEnv(TorchVecEnv):
    """Torch port of ``LunarLanderVecEnv`` (identical dynamics/rewards)."""

    max_episode_steps = 1000

    DT = 0.02
    GRAVITY = 10.0
    MAIN_ACCEL = 15.0
    SIDE_ACCEL = 1.5
    SIDE_TORQUE = 3.0
    ANGLE_DAMP = 0.99
    X_WORLD = 10.0
    Y_WORLD = 20.0 / 3.0
    LEG_Y = 0.1

    def __init__(self, num_envs: int = 1, device: str = "cuda", seed: Optional[int] = None):
        super().__init__(num_envs, device, seed)
        high = np.full(8, np.inf, dtype=np.float32)
        self.single_observation_space = Box(-high, high)
        self.single_action_space = Discrete(4)
        N = self.num_envs
        self.state = torch.zeros(N, 6, device=device)
        self.legs = torch.zeros(N, device=device)
        self.prev_shaping = torch.zeros(N, device=device)

    def _reset_rows(self, mask: torch.Tensor) -> None:
        N = self.num_envs
        s = torch.empty(N, 6, device=self.device)
        s[:, 0] = self._rand(N, low=-0.5, high=0.5)
        s[:, 1] = 10.0
        s[:, 2] = self._rand(N, low=-2.0, high=2.0)
        s[:, 3] = self._rand(N, low=-1.5, high=0.5)
        s[:, 4] = self._rand(N, low=-0.15, high=0.15)
        s[:, 5] = self._rand(N, low=-0.3, high=0.3)
        m = mask.unsqueeze(1)
        self.state.copy_(torch.where(m, s, self.state))
        self.legs.copy_(torch.where(mask, torch.zeros_like(self.legs), self.legs))
        self.prev_shaping.copy_(torch.where(mask, self._shaping(), self.prev_shaping))

    def _obs(self) -> torch.Tensor:
        px, py, vx, vy, ang, vang = self.state.unbind(1)
        return torch.stack(
            [px / self.X_WORLD, py / self.Y_WORLD, vx / 5.0, vy / 7.5, ang, 0.4 * vang,
             self.legs, self.legs],
            dim=1,
        )

    def _shaping(self) -> torch.Tensor:
        o = self._obs()
        return (
            -100.0 * torch.sqrt(o[:, 0] ** 2 + o[:, 1] ** 2)
            - 100.0 * torch.sqrt(o[:, 2] ** 2 + o[:, 3] ** 2)
            - 100.0 * o[:, 4].abs()
            + 20.0 * o[:, 6]
        )

    def _step_all(self, actions: torch.Tensor):
        a = actions.reshape(-1).long()
        px, py, vx, vy, ang, vang = self.state.unbind(1)
        main = (a == 2).float()
        left = (a == 1).float()
        right = (a == 3).float()

        sin_a, cos_a = torch.sin(ang), torch.cos(ang)
        ax = -sin_a * self.MAIN_ACCEL * main + (right - left) * self.SIDE_ACCEL * cos_a
        ay = cos_a * self.MAIN_ACCEL * main - self.GRAVITY + (right - left) * self.SIDE_ACCEL * sin_a
        aang = (left - right) * self.SIDE_TORQUE

        vx = vx + ax * self.DT
        vy = vy + ay * self.DT
        vang = (vang + aang * self.DT) * self.ANGLE_DAMP
        px = px + vx * self.DT
        py = py + vy * self.DT
        ang = ang + vang * self.DT

        on_ground = py <= self.LEG_Y
        upright = ang.abs() < 0.4
        self.legs.copy_((on_ground & upright).float())
        py = py.clamp(min=0.0)
        grounded = py <= 1e-9
        impact_speed = torch.sqrt(vx**2 + vy**2)

        self.state.copy_(torch.stack([px, py, vx, vy, ang, vang], dim=1))
        shaping = self._shaping()
        reward = shaping - self.prev_shaping
        self.prev_shaping.copy_(shaping)
        reward = reward - (0.30 * main + 0.03 * (left + right))

        vy = torch.where(grounded & (vy < 0), torch.zeros_like(vy), vy)
        vx = torch.where(grounded, vx * 0.8, vx)
        vang = torch.where(grounded, vang * 0.5, vang)
        self.state.copy_(torch.stack([px, py, vx, vy, ang, vang], dim=1))

        crash = (grounded & (~upright | (impact_speed > 1.5))) | (px.abs() > self.X_WORLD)
        landed = grounded & upright & (impact_speed <= 1.5) & (vang.abs() < 0.3)
        terminated = crash | landed
        reward = torch.where(crash, reward - 100.0, reward)
        reward = torch.where(landed, reward + 100.0, reward)
        return reward, terminated


class CatchPongTorchVecEnv(TorchVecEnv):
    """Device-resident port of ``CatchPongVecEnv`` (uint8 frame stacks stay
    in HBM; the Rainbow pipeline never crosses PCIe for observations)."""

    max_episode_steps = 2000
    H = W = 84
    FRAMES = 4
    PADDLE_W = 12
    BALL = 3

    def __init__(self, num_envs: int = 1, device: str = "cuda", seed: Optional[int] = None):
        super().__init__(num_envs, device, seed)
        self.single_observation_space = Box(0, 255, (self.FRAMES, self.H, self.W), dtype=np.uint8)
        self.single_action_space = Discrete(3)
        N = self.num_envs
        self.ball = torch.zeros(N, 3, device=device)  # x, y, vx
        self.paddle_x = torch.zeros(N, device=device)
        self.frames = torch.zeros(N, self.FRAMES, self.H, self.W, dtype=torch.uint8, device=device)
        self._rows = torch.arange(N, device=device)
        self._ball_off = torch.stack(
            torch.meshgrid(torch.arange(self.BALL, device=device),
                           torch.arange(self.BALL, device=device), indexing="ij"),
            dim=-1,
        ).reshape(-1, 2)  # (9, 2) dy,dx
        self._pad_off = torch.arange(self.PADDLE_W, device=device)

    def _spawn_ball(self, mask: torch.Tensor) -> None:
        N = self.num_envs
        new = torch.stack(
            [self._rand(N, low=5.0, high=self.W - 5.0),
             torch.full((N,), 2.0, device=self.device),
             self._rand(N, low=-1.0, high=1.0)], dim=1)
        self.ball.copy_(torch.where(mask.unsqueeze(1), new, self.ball))

    def _reset_rows(self, mask: torch.Tensor) -> None:
        self._spawn_ball(mask)
        self.paddle_x.copy_(torch.where(mask, torch.full_like(self.paddle_x, self.W / 2), self.paddle_x))
        self.frames.copy_(torch.where(mask.view(-1, 1, 1, 1), torch.zeros_like(self.frames), self.frames))
        self._render()

    def _render(self) -> None:
        self.frames[:, :-1].copy_(self.frames[:, 1:].clone())
        frame = torch.zeros(self.num_envs, self.H, self.W, dtype=torch.uint8, device=self.device)
        bx = self.ball[:, 0].long().clamp(0, self.W - self.BALL)
        by = self.ball[:, 1].long().clamp(0, self.H - self.BALL)
        px = self.paddle_x.long().clamp(0, self.W - self.PADDLE_W)
        rows = self._rows.view(-1, 1)
        # ball: 9 pixels per env via flat index_put
        yy = (by.view(-1, 1) + self._ball_off[:, 0].view(1, -1)).reshape(-1)
        xx = (bx.view(-1, 1) + self._ball_off[:, 1].view(1, -1)).reshape(-1)
        rr = rows.expand(-1, self._ball_off.shape[0]).reshape(-1)
        frame[rr, yy, xx] = 255
        cols = (px.view(-1, 1) + self._pad_off.view(1, -1))
        frame[rows.expand_as(cols), self.H - 3, cols] = 180
        frame[rows.expand_as(cols), self.H - 2, cols] = 180
        self.frames[:, -1].copy_(frame)

    def _obs(self) -> torch.Tensor:
        return self.frames.clone()

    def _step_all(self, actions: torch.Tensor):
        a = actions.reshape(-1).long()
        self.paddle_x.add_(torch.where(a == 1, -3.0, torch.where(a == 2, 3.0, torch.zeros_like(self.paddle_x))))
        self.paddle_x.clamp_(0, self.W - self.PADDLE_W)
        bx, by, bvx = self.ball.unbind(1)
        by = by + 2.0
        bx = bx + bvx
        bounce = (bx <= 0) | (bx >= self.W - self.BALL)
        bvx = torch.where(bounce, -bvx, bvx)
        bx = bx.clamp(0, float(self.W - self.BALL))
        self.ball.copy_(torch.stack([bx, by, bvx], dim=1))

        at_bottom = by >= self.H - 5
        caught = at_bottom & (bx + self.BALL >= self.paddle_x) & (bx <= self.paddle_x + self.PADDLE_W)
        missed = at_bottom & ~caught
        reward = torch.where(caught, 1.0, torch.where(missed, -1.0, torch.zeros_like(bx)))
        self._spawn_ball(at_bottom)
        self._render()
        return reward, missed


class CartPoleTorchVecEnv(TorchVecEnv):
    """Torch port of ``CartPoleVecEnv``."""

    max_episode_steps = 500

    def __init__(self, num_envs: int = 1, device: str = "cuda", seed: Optional[int] = None):
        super().__init__(num_envs, device, seed)
        high = np.array([4.8, np.finfo(np.float32).max, 0.42, np.finfo(np.float32).max], dtype=np.float32)
        self.single_observation_space = Box(-high, high)
        self.single_action_space = Discrete(2)
        self.state = torch.zeros(num_envs, 4, device=device)

    def _reset_rows(self, mask: torch.Tensor) -> None:
        s = self._rand(self.num_envs, 4, low=-0.05, high=0.05)
        self.state.copy_(torch.where(mask.unsqueeze(1), s, self.state))

    def _obs(self) -> torch.Tensor:
        return self.state.clone()

    def _step_all(self, actions: torch.Tensor):
        x, x_dot, theta, theta_dot = self.state.unbind(1)
        force = torch.where(actions.reshape(-1) == 1, 10.0, -10.0)
        costheta, sintheta = torch.cos(theta), torch.sin(theta)
        temp = (force + 0.05 * theta_dot**2 * sintheta) / 1.1
        thetaacc = (9.8 * sintheta - costheta * temp) / (0.5 * (4.0 / 3.0 - 0.1 * costheta**2 / 1.1))
        xacc = temp - 0.05 * thetaacc * costheta / 1.1
        x = x + 0.02 * x_dot
        x_dot = x_dot + 0.02 * xacc
        theta = theta + 0.02 * theta_dot
        theta_dot = theta_dot + 0.02 * thetaacc
        self.state.copy_(torch.stack([x, x_dot, theta, theta_dot], dim=1))
        terminated = (x.abs() > 2.4) | (theta.abs() > 12 * 3.14159 / 180)
        return torch.ones(self.num_envs, device=self.device), terminated
