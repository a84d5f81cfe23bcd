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

    def __init__(self, *args, vf_coef: float = 0.5, value_head_lr: Optional[float] = None, **kwargs):
        kwargs.setdefault("group_size", 1)
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

        B = ids.shape[0]
        mb = max(self.micro_batch_size, 1)

        # value baseline + old logprobs
        with torch.no_grad():
            values = torch.empty(B, device=self.device)
            old_logp = torch.empty(action_mask.shape, device=self.device)
            ref_logp = torch.empty_like(old_logp) if self.beta > 0 else None
            for s in range(0, B, mb):
                e = min(s + mb, B)
                values[s:e] = self._sequence_values(ids[s:e], attention_mask[s:e])
                old_logp[s:e] = self.compute_logprobs(ids[s:e], attention_mask[s:e])
                if ref_logp is not None:
                    ref_logp[s:e] = self.compute_logprobs(ids[s:e], attention_mask[s:e], adapter=None)

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
                # value head regression toward realized reward
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
