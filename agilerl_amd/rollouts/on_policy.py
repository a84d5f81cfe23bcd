"""Vectorized on-policy rollout collection.

Reference parity: ``agilerl/rollouts/on_policy.py`` (collect_rollouts :221,
collect_rollouts_recurrent :243; shared ``_collect_rollouts`` :29).

The buffer lives on the agent's device; env stepping stays on host numpy
and observations stream to HBM batched (pinned staging handled by the
buffer).  Truncation bootstrapping uses ``info["final_observation"]``.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import numpy as np
import torch

from ..components.rollout_buffer import RolloutBuffer

__all__ = ["collect_rollouts", "collect_rollouts_device", "collect_rollouts_recurrent", "collect_rollouts_llm"]


def collect_rollouts(
    agent,
    env,
    buffer: RolloutBuffer,
    n_steps: int,
    obs: Optional[np.ndarray] = None,
    done: Optional[np.ndarray] = None,
) -> Tuple[np.ndarray, np.ndarray, Dict[str, float]]:
    """Collect ``n_steps`` vectorized steps into ``buffer``; computes GAE.

    Returns (last_obs, last_done, info_stats).
    """
    if obs is None:
        obs, _ = env.reset()
        done = np.zeros(env.num_envs, dtype=bool)
    ep_returns: list = []
    buffer.reset()
    for _ in range(n_steps):
        action, log_prob, _entropy, value = agent.get_action(obs, training=True)
        next_obs, reward, term, trunc, info = env.step(action)
        # bootstrap through truncation: add V(final_obs) to the reward
        if np.any(trunc) and "final_observation" in info:
            with torch.no_grad():
                v_final = agent.get_values(info["final_observation"]).cpu().numpy()
            reward = np.where(trunc, reward + agent.gamma * v_final, reward)
        buffer.add(
            obs=obs,
            action=action,
            reward=reward,
            done=(term | trunc).astype(np.float32),
            value=value,
            log_prob=log_prob,
        )
        obs = next_obs
        done = term | trunc
        if "episode_return" in info:
            ep_returns.extend(np.asarray(info["episode_return"]).tolist())
    last_value = agent.get_values(obs)
    buffer.compute_returns_and_advantages(last_value, torch.as_tensor(done, dtype=torch.float32))
    stats = {"mean_episode_return": float(np.mean(ep_returns))} if ep_returns else {}
    return obs, done, stats


def collect_rollouts_device(
    agent,
    env,
    buffer: RolloutBuffer,
    n_steps: int,
    obs: Optional[torch.Tensor] = None,
    done: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor, Dict[str, float]]:
    """Device-native collection for ``TorchVecEnv`` (obs/actions/rewards
    never leave HBM; one host sync per rollout for the episode stats)."""
    if obs is None:
        obs, _ = env.reset()
        done = torch.zeros(env.num_envs, dtype=torch.bool, device=obs.device)
    buffer.reset()
    ep_sum = torch.zeros((), device=obs.device)
    ep_cnt = torch.zeros((), device=obs.device)
    for _ in range(n_steps):
        action, log_prob, _entropy, value = agent.get_action_device(obs)
        next_obs, reward, term, trunc, info = env.step(action)
        # branchless truncation bootstrap: V(final_obs) only credited on trunc rows
        v_final = agent.get_values(info["final_observation"])
        reward = torch.where(trunc, reward + agent.gamma * v_final, reward)
        done_t = info["done_mask"]
        buffer.add(
            obs=obs,
            action=action,
            reward=reward,
            done=done_t.float(),
            value=value,
            log_prob=log_prob,
        )
        ep_sum = ep_sum + (info["episode_return"] * done_t.float()).sum()
        ep_cnt = ep_cnt + done_t.float().sum()
        obs = next_obs
        done = done_t
    last_value = agent.get_values(obs)
    buffer.compute_returns_and_advantages(last_value, done.float())
    stats = {}
    cnt = float(ep_cnt)
    if cnt > 0:
        stats["mean_episode_return"] = float(ep_sum) / cnt
    return obs, done, stats


def collect_rollouts_recurrent(
    agent,
    env,
    buffer: RolloutBuffer,
    n_steps: int,
    obs: Optional[np.ndarray] = None,
    done: Optional[np.ndarray] = None,
    hidden: Optional[Dict[str, torch.Tensor]] = None,
):
    """Recurrent variant: threads LSTM hidden state through collection and
    stores the per-step hidden for BPTT sequence minibatches (reference
    rollouts/on_policy.py:243).  Hidden is zeroed for finished env rows."""
    if obs is None:
        obs, _ = env.reset()
        done = np.zeros(env.num_envs, dtype=bool)
        hidden = agent.init_hidden(env.num_envs)
    buffer.reset()
    ep_returns: list = []
    for _ in range(n_steps):
        # store pre-step hidden (B, L, H) so sequences can rebuild h0
        stored_hidden = {k: v.transpose(0, 1).contiguous() for k, v in hidden.items()}
        action, log_prob, value, hidden = agent.get_action_recurrent(obs, hidden)
        next_obs, reward, term, trunc, info = env.step(action)
        done_now = term | trunc
        buffer.add(
            obs=obs,
            action=action,
            reward=reward,
            done=done_now.astype(np.float32),
            value=value,
            log_prob=log_prob,
            hidden_state=stored_hidden,
        )
        # zero hidden rows for finished episodes
        if done_now.any():
            mask = torch.as_tensor(~done_now, dtype=torch.float32, device=hidden["ha"].device)
            hidden = {k: v * mask.view(1, -1, 1) for k, v in hidden.items()}
        obs = next_obs
        done = done_now
        if "episode_return" in info:
            ep_returns.extend(np.asarray(info["episode_return"]).tolist())
    with torch.no_grad():
        v_out, _ = agent.critic.forward_step(obs, (hidden["hc"], hidden["cc"]))
    buffer.compute_returns_and_advantages(
        v_out.squeeze(-1), torch.as_tensor(done, dtype=torch.float32)
    )
    stats = {"mean_episode_return": float(np.mean(ep_returns))} if ep_returns else {}
    return obs, done, hidden, stats


def collect_rollouts_llm(agent, env, n_batches: int = 1):
    """One or more LLM prompt-batch rollouts (reference
    rollouts/on_policy.py:265 collect_rollouts_llm): reset -> generate ->
    score -> experiences dict ready for ``agent.learn``.

    Returns (experiences, mean_reward) for the LAST batch when
    ``n_batches == 1`` (the common case), else a list of experience
    dicts.  Sampling logprobs/turn metadata captured by the paged engine
    or multiturn envs ride inside the experiences.
    """
    from ..llm_envs.base import make_grpo_experiences

    pad_id = getattr(agent.model.config, "pad_token_id", None) or 0
    out = []
    for _ in range(max(int(n_batches), 1)):
        prompts = env.reset()
        sequences = agent.get_action(prompts, training=True)
        rewards = env.score(sequences)
        experiences = make_grpo_experiences(env, sequences, rewards, pad_token_id=pad_id)
        sampling = getattr(agent, "last_sampling_logps", None)
        if sampling is not None and "sampling_logps" not in experiences:
            experiences["sampling_logps"] = sampling
        out.append((experiences, float(np.mean(rewards))))
    return out[0] if len(out) == 1 else out
