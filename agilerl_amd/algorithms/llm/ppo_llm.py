"""LLM PPO: clipped policy loss with a learned value-head baseline.

Reference parity: ``agilerl/algorithms/ppo_llm.py:69`` (value head via
``utils/ppo_value_head.py``, turn-level GAE :1141; single-turn here:
per-sequence value from the last prompt-token hidden state, advantage =
reward - V, token-level clipped surrogate via the fused kernel).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch
import torch.nn as nn

from .grpo import GRPO

__all__ = ["PPOLLM"]


class PPOLLM(GRPO):
    """GRPO machinery with a value-head baseline instead of group centering."""

    def __init__(self, *args, vf_coef: float = 0.5, value_head_lr: Optional[float] = None,
                 gamma: float = 1.0, gae_lambda: float = 1.0, **kwargs):
        kwargs.setdefault("group_size", 1)
        self.gamma = float(gamma)
        self.gae_lambda = float(gae_lambda)
        # reference ppo_llm.py multi-turn kwargs: turn_ratio_pooling pools
        # the IS ratio per turn (our importance_sampling_level="turn");
        # turn_level_clip clips at that pooled level (implied by the level);
        # turn_value_reduction is vLLM-era value bookkeeping (ignored)
        if kwargs.pop("turn_ratio_pooling", False) or kwargs.pop("turn_level_clip", False):
            kwargs.setdefault("importance_sampling_level", "turn")
        if kwargs.pop("turn_value_reduction", None) is not None:
            import warnings

            warnings.warn("turn_value_reduction ignored: the value head pools "
                          "at the last prompt token", RuntimeWarning)
        if "lr_actor" in kwargs:
            kwargs.setdefault("lr", kwargs.pop("lr_actor"))
        if "lr_critic" in kwargs:
            value_head_lr = value_head_lr or kwargs.pop("lr_critic")
        else:
            kwargs.pop("lr_critic", None)
        super().__init__(*args, **kwargs)
        self.algo = "PPOLLM"
        self.vf_coef = float(vf_coef)
        hidden = self.model.config.hidden_size
        self.value_head = nn.Linear(hidden, 1).to(self.device)
        self.value_head.weight.data.normal_(0, 0.02)
        self.value_head.bias.data.zero_()
        self.value_optimizer = torch.optim.AdamW(
            self.value_head.parameters(), lr=value_head_lr or self.lr * 10
        )

    # ------------------------------------------------------------------
    def _sequence_values(self, ids, attention_mask, with_grad: bool = False) -> torch.Tensor:
        """V(prompt): value head on the hidden state at the last prompt token
        (= first completion position - 1)."""
        self._activate("self")
        ctx = torch.enable_grad() if with_grad else torch.no_grad()
        with ctx:
            hidden = self._decoder()(
                input_ids=ids, attention_mask=attention_mask
            ).last_hidden_state
            # last non-pad position per row
            lengths = attention_mask.sum(dim=1).clamp(min=1) - 1
            pooled = hidden[torch.arange(ids.shape[0], device=ids.device), lengths]
            return self.value_head(pooled.float()).squeeze(-1)

    def _calculate_advantages(self, rewards: torch.Tensor) -> torch.Tensor:
        # placeholder — learn() computes advantage with the value baseline
        return rewards

    def _turn_values(self, ids, attention_mask, turn_ids, with_grad: bool = False,
                     num_turns=None):
        """V per turn: value head on the hidden state at each turn's last
        token (reference ppo_llm.py turn-level critic).  turn_ids indexes
        TARGETS (position j predicts ids[j+1]), so turn k's last hidden
        state sits at ids position (last j with turn_ids==k) + 1."""
        self._activate("self")
        B = ids.shape[0]
        K = int(num_turns) if num_turns is not None else int(turn_ids.max().item()) + 1
        ctx = torch.enable_grad() if with_grad else torch.no_grad()
        with ctx:
            hidden = self._decoder()(
                input_ids=ids, attention_mask=attention_mask
            ).last_hidden_state
            T1 = turn_ids.shape[1]
            pos = torch.arange(T1, device=ids.device).view(1, 1, T1)
            is_turn = turn_ids.unsqueeze(1) == torch.arange(K, device=ids.device).view(1, K, 1)
            # (B, K): last target index of each turn (-1 if unplayed)
            last = torch.where(is_turn, pos, torch.full_like(pos, -1)).amax(dim=2)
            played = last >= 0
            gather_pos = (last.clamp(min=0) + 1).clamp(max=ids.shape[1] - 1)
            pooled = hidden[torch.arange(B, device=ids.device).unsqueeze(1), gather_pos]
            vals = self.value_head(pooled.float()).squeeze(-1)
            return vals * played.float(), played

    def _turn_gae(self, turn_rewards, turn_values, played):
        """GAE across turns (reference ppo_llm.py:1141): gamma discounts
        BETWEEN turns, not within; unplayed turns carry nothing."""
        B, K = turn_rewards.shape
        n_turns = played.float().sum(dim=1)  # per-sample played count
        adv = torch.zeros_like(turn_rewards)
        last_gae = torch.zeros(B, device=turn_rewards.device)
        for t in reversed(range(K)):
            is_last = (t >= n_turns - 1)
            next_v = turn_values[:, t + 1] if t < K - 1 else torch.zeros(B, device=turn_rewards.device)
            next_v = torch.where(is_last, torch.zeros_like(next_v), next_v)
            delta = turn_rewards[:, t] + self.gamma * next_v - turn_values[:, t]
            has = (n_turns > t).float()
            last_gae = (delta + self.gamma * self.gae_lambda * last_gae) * has
            adv[:, t] = last_gae
        return adv

    def learn(self, experiences: Dict[str, Any]) -> Dict[str, float]:
        ids = experiences["ids"].to(self.device)
        attention_mask = experiences.get("attention_mask")
        attention_mask = (
            torch.ones_like(ids) if attention_mask is None else attention_mask.to(self.device)
        )
        action_mask = experiences["action_mask"].to(self.device).float()
        rewards = experiences["rewards"].to(self.device).float()
        sampling_logps = experiences.get("sampling_logps")
        if sampling_logps is not None and self.sampling_is_correction:
            sampling_logps = sampling_logps.to(self.device).float()
        else:
            sampling_logps = None

        turn_ids = experiences.get("turn_ids")
        turn_rewards = experiences.get("turn_rewards")
        multi_turn = turn_ids is not None and turn_rewards is not None
        if multi_turn:
            turn_ids = turn_ids.to(self.device)
            turn_rewards = turn_rewards.to(self.device).float()

        B = ids.shape[0]
        mb = max(self.micro_batch_size, 1)

        # value baseline + old logprobs
        with torch.no_grad():
            values = torch.empty(B, device=self.device)
            old_logp = torch.empty(action_mask.shape, device=self.device)
            ref_logp = torch.empty_like(old_logp) if self.beta > 0 else None
            tv = played = None
            if multi_turn:
                K = int(turn_ids.max().item()) + 1
                tv = torch.empty(B, K, device=self.device)
                played = torch.empty(B, K, dtype=torch.bool, device=self.device)
            for s in range(0, B, mb):
                e = min(s + mb, B)
                if multi_turn:
                    tv[s:e], played[s:e] = self._turn_values(
                        ids[s:e], attention_mask[s:e], turn_ids[s:e], num_turns=K
                    )
                else:
                    values[s:e] = self._sequence_values(ids[s:e], attention_mask[s:e])
                old_logp[s:e] = self.compute_logprobs(ids[s:e], attention_mask[s:e])
                if ref_logp is not None:
                    ref_logp[s:e] = self.compute_logprobs(ids[s:e], attention_mask[s:e], adapter=None)

        if multi_turn:
            # reference ppo_llm.py:1141 turn-level GAE: each turn is one RL
            # action; gamma/gae_lambda act across turns
            K = tv.shape[1]
            tr = turn_rewards[:, :K] if turn_rewards.shape[1] >= K else torch.nn.functional.pad(
                turn_rewards, (0, K - turn_rewards.shape[1]))
            turn_adv = self._turn_gae(tr, tv, played)
            turn_returns = turn_adv + tv
            flat = turn_adv[played]
            if flat.numel() > 1:
                turn_adv = torch.where(
                    played, (turn_adv - flat.mean()) / (flat.std() + 1e-8),
                    torch.zeros_like(turn_adv))
            safe = turn_ids.clamp(min=0, max=K - 1).long()
            adv_tok = turn_adv.gather(1, safe) * (turn_ids >= 0).float() * action_mask
        else:
            advantages = rewards - values
            if advantages.numel() > 1:
                advantages = (advantages - advantages.mean()) / (advantages.std() + 1e-8)
            adv_tok = advantages.unsqueeze(1).expand_as(action_mask)

        clip_hi, clip_lo = 1.0 + self.clip_coef, 1.0 - self.clip_coef
        stats = {"loss": 0.0, "value_loss": 0.0}
        n = 0
        for _ in range(self.update_epochs):
            perm = torch.randperm(B, device=self.device)
            for s in range(0, B, mb):
                sel = perm[s : s + mb]
                logp = self.compute_logprobs(ids[sel], attention_mask[sel], with_grad=True)
                loss = self._policy_loss(
                    logp, old_logp[sel], adv_tok[sel], action_mask[sel],
                    ref_logp[sel] if ref_logp is not None else None, clip_lo, clip_hi,
                    sampling_logp=sampling_logps[sel] if sampling_logps is not None else None,
                )
                self.backward_and_step(loss)
                # value head regression toward realized reward / turn returns
                if multi_turn:
                    v_sel, p_sel = self._turn_values(
                        ids[sel], attention_mask[sel], turn_ids[sel], with_grad=True,
                        num_turns=turn_returns.shape[1],
                    )
                    tgt = turn_returns[sel]
                    pf = p_sel.float()
                    v_loss = self.vf_coef * (
                        ((v_sel - tgt) ** 2 * pf).sum() / pf.sum().clamp(min=1.0)
                    )
                else:
                    v = self._sequence_values(ids[sel], attention_mask[sel], with_grad=True)
                    v_loss = self.vf_coef * ((v - rewards[sel]) ** 2).mean()
                self.value_optimizer.zero_grad()
                v_loss.backward()
                self.value_optimizer.step()
                stats["loss"] += float(loss.detach())
                stats["value_loss"] += float(v_loss.detach())
                n += 1
        stats = {k: v / max(n, 1) for k, v in stats.items()}
        stats["mean_reward"] = float(rewards.mean())
        return stats

    # value head participates in clone/checkpoint
    def clone(self, index: Optional[int] = None, wrap: bool = True):
        clone = super().clone(index=index, wrap=wrap)
        clone.value_head = nn.Linear(self.value_head.in_features, 1).to(self.device)
        clone.value_head.load_state_dict(self.value_head.state_dict())
        clone.value_optimizer = torch.optim.AdamW(
            clone.value_head.parameters(), lr=self.value_optimizer.param_groups[0]["lr"]
        )
        return clone
