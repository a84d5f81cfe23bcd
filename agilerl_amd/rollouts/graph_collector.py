"""hipGraph-captured rollout collection (PPO, discrete actions).

The device-native collect loop is launch-bound: one iteration =
actor forward + sample + critic forward + env physics + buffer writes
~= 60-100 tiny kernel launches, ~6 ms of CPU dispatch per iteration on
MI355X (profiles/r01_notes.md).  This module captures ONE whole
iteration into a hipGraph (torch.cuda.CUDAGraph == hipGraph on ROCm) and
replays it ``n_steps`` times — dispatch cost collapses to one replay
per env step.

Requirements engineered into the stack for this:
- ``TorchVecEnv`` steps are branchless with IN-PLACE state updates
  (stable addresses across replays).
- Sampling is Gumbel-argmax (explicit ``torch.rand``) instead of
  ``Categorical.sample`` — capture-safe RNG, fixed shapes.
- Rollout storage is pre-allocated (T, N, ...); the write index is a
  device tensor read by ``index_copy_`` inside the graph.

Re-capture is needed after architecture mutations (module objects
change); weight updates from the optimizer are visible to replays
because parameters update in place.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
import torch.nn.functional as F

from .. import ops

__all__ = ["GraphedPPOCollector"]


class GraphedPPOCollector:
    def __init__(self, agent, env, n_steps: int):
        self.agent = agent
        self.env = env
        self.n_steps = int(n_steps)
        self.device = env.device
        N = env.num_envs
        obs_dim = env.single_observation_space.shape[0]
        dev = self.device

        T = self.n_steps
        self.storage = {
            "obs": torch.zeros(T, N, obs_dim, device=dev),
            "action": torch.zeros(T, N, dtype=torch.long, device=dev),
            "reward": torch.zeros(T, N, device=dev),
            "done": torch.zeros(T, N, device=dev),
            "value": torch.zeros(T, N, device=dev),
            "log_prob": torch.zeros(T, N, device=dev),
        }
        self.pos = torch.zeros(1, dtype=torch.long, device=dev)
        self.obs = torch.zeros(N, obs_dim, device=dev)
        self.done = torch.zeros(N, device=dev)
        self.ep_sum = torch.zeros(1, device=dev)
        self.ep_cnt = torch.zeros(1, device=dev)
        self.graph: Optional[torch.cuda.CUDAGraph] = None

    # ------------------------------------------------------------------
    def _iteration(self) -> None:
        """One capture-safe collect iteration reading/writing static state."""
        agent, env = self.agent, self.env
        obs = self.obs
        logits = agent.actor(agent.actor.preprocess(obs))
        # Gumbel-argmax sampling (graph-safe RNG)
        u = torch.rand_like(logits).clamp_(1e-10, 1.0)
        gumbel = -torch.log(-torch.log(u))
        action = (logits + gumbel).argmax(dim=-1)
        log_probs = F.log_softmax(logits, dim=-1)
        log_prob = log_probs.gather(1, action.unsqueeze(1)).squeeze(1)
        value = agent.critic(agent.critic.preprocess(obs)).squeeze(-1)

        next_obs, reward, term, trunc, info = env.step(action)
        v_final = agent.critic(agent.critic.preprocess(info["final_observation"])).squeeze(-1)
        reward = torch.where(trunc, reward + agent.gamma * v_final, reward)
        done_t = info["done_mask"].float()

        idx = self.pos
        self.storage["obs"].index_copy_(0, idx, obs.unsqueeze(0))
        self.storage["action"].index_copy_(0, idx, action.unsqueeze(0))
        self.storage["reward"].index_copy_(0, idx, reward.unsqueeze(0))
        self.storage["done"].index_copy_(0, idx, done_t.unsqueeze(0))
        self.storage["value"].index_copy_(0, idx, value.unsqueeze(0))
        self.storage["log_prob"].index_copy_(0, idx, log_prob.unsqueeze(0))
        self.pos.add_(1)

        self.ep_sum.add_((info["episode_return"] * done_t).sum().reshape(1))
        self.ep_cnt.add_(done_t.sum().reshape(1))
        self.obs.copy_(next_obs)
        self.done.copy_(done_t)

    # ------------------------------------------------------------------
    def capture(self) -> None:
        """Warm up and capture the iteration graph."""
        env = self.env
        if hasattr(env, "gen"):
            env.graph_safe = True
        obs0, _ = env.reset()
        self.obs.copy_(obs0)
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.no_grad(), torch.cuda.stream(s):
            for _ in range(3):  # warmup allocations on the side stream
                self._iteration()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.pos.zero_()
        self.graph = torch.cuda.CUDAGraph()
        with torch.no_grad(), torch.cuda.graph(self.graph):
            self._iteration()
        # capture ran one iteration symbolically; reset counters
        self.pos.zero_()
        self.ep_sum.zero_()
        self.ep_cnt.zero_()
        obs0, _ = env.reset()
        self.obs.copy_(obs0)
        self.done.zero_()

    # ------------------------------------------------------------------
    @torch.no_grad()
    def collect(self) -> Tuple[Dict[str, torch.Tensor], Dict[str, float]]:
        """Replay the graph n_steps times; returns (flat rollout dict with
        advantages/returns, stats)."""
        if self.graph is None:
            self.capture()
        self.pos.zero_()
        self.ep_sum.zero_()
        self.ep_cnt.zero_()
        for _ in range(self.n_steps):
            self.graph.replay()
        agent = self.agent
        last_value = agent.critic(agent.critic.preprocess(self.obs)).squeeze(-1)
        adv, ret = ops.gae_scan(
            self.storage["reward"], self.storage["value"], self.storage["done"],
            last_value, agent.gamma, agent.gae_lambda,
        )
        T, N = self.storage["reward"].shape
        # clone: the storage is reused by the next collect; returning views
        # would silently mutate a held rollout (caught by the aliasing test)
        flat = {k: v.reshape(T * N, *v.shape[2:]).clone() for k, v in self.storage.items()}
        flat["advantages"] = adv.reshape(-1)
        flat["returns"] = ret.reshape(-1)
        stats = {}
        cnt = float(self.ep_cnt)
        if cnt > 0:
            stats["mean_episode_return"] = float(self.ep_sum) / cnt
        return flat, stats
