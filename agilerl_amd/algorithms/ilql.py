"""ILQL — implicit language Q-learning (offline token-level RL).

Reference parity: ``agilerl/algorithms/ilql.py:42`` (legacy stack on
EvolvableGPT: per-token twin Q + V heads, expectile V regression, CQL
regularizer, polyak target heads, beta-perturbed decoding).  Compact
re-implementation on this framework's :class:`EvolvableGPT`.
"""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..modules.gpt import EvolvableGPT

__all__ = ["ILQL", "ILQL_Policy", "ILQL_Evaluator"]


class ILQL(nn.Module):
    def __init__(
        self,
        vocab_size: int,
        n_layer: int = 4,
        n_head: int = 4,
        n_embd: int = 128,
        max_positions: int = 512,
        gamma: float = 0.99,
        tau: float = 0.7,           # expectile
        cql_weight: float = 0.01,
        awac_beta: float = 4.0,     # advantage-weighted CE temperature
        polyak: float = 5e-3,
        lr: float = 1e-4,
        device: str = "cpu",
    ):
        super().__init__()
        self.vocab_size = vocab_size
        self.gamma = gamma
        self.tau = tau
        self.cql_weight = cql_weight
        self.awac_beta = awac_beta
        self.polyak = polyak
        self.device = device

        self.gpt = EvolvableGPT(
            vocab_size, n_layer=n_layer, n_head=n_head, n_embd=n_embd,
            max_positions=max_positions, device=device,
        )
        H = self.gpt.n_embd
        self.q1_head = nn.Linear(H, vocab_size).to(device)
        self.q2_head = nn.Linear(H, vocab_size).to(device)
        self.v_head = nn.Linear(H, 1).to(device)
        self.q1_target = nn.Linear(H, vocab_size).to(device)
        self.q2_target = nn.Linear(H, vocab_size).to(device)
        self.q1_target.load_state_dict(self.q1_head.state_dict())
        self.q2_target.load_state_dict(self.q2_head.state_dict())
        for p in list(self.q1_target.parameters()) + list(self.q2_target.parameters()):
            p.requires_grad = False
        self.optimizer = torch.optim.AdamW(
            [p for p in self.parameters() if p.requires_grad], lr=lr
        )

    # ------------------------------------------------------------------
    def learn(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        """batch: ids (B, T), rewards (B, T-1) per-target-token rewards,
        mask (B, T-1) valid target positions."""
        ids = batch["ids"].to(self.device)
        rewards = batch["rewards"].to(self.device).float()
        mask = batch["mask"].to(self.device).float()
        B, T = ids.shape

        hidden = self.gpt.transformer_forward(ids)  # (B, T, H)
        h_in = hidden[:, :-1]                        # predicts token t+1
        actions = ids[:, 1:]                         # taken "actions"

        q1 = self.q1_head(h_in)
        q2 = self.q2_head(h_in)
        v = self.v_head(hidden).squeeze(-1)          # V(s_t) for all t
        q1_a = q1.gather(-1, actions.unsqueeze(-1)).squeeze(-1)
        q2_a = q2.gather(-1, actions.unsqueeze(-1)).squeeze(-1)

        with torch.no_grad():
            tq1 = self.q1_target(h_in).gather(-1, actions.unsqueeze(-1)).squeeze(-1)
            tq2 = self.q2_target(h_in).gather(-1, actions.unsqueeze(-1)).squeeze(-1)
            tq = torch.minimum(tq1, tq2)
            # V target for the Bellman backup: V(s_{t+1}); terminal at T-1
            v_next = torch.cat([v[:, 2:], torch.zeros(B, 1, device=v.device)], dim=1).detach()
            q_target = rewards + self.gamma * v_next

        # expectile V loss toward target-Q at the taken action
        diff = tq - v[:, :-1]
        w = torch.where(diff > 0, self.tau, 1 - self.tau)
        v_loss = ops.masked_mean(w * diff**2, mask)

        q_loss = ops.masked_mean((q1_a - q_target) ** 2 + (q2_a - q_target) ** 2, mask)
        cql = ops.masked_mean(
            torch.logsumexp(q1, dim=-1) - q1_a + torch.logsumexp(q2, dim=-1) - q2_a, mask
        )
        # advantage-weighted CE on the LM head (policy extraction)
        logits = self.gpt.model["head"](h_in)
        with torch.no_grad():
            adv = (tq - v[:, :-1]).clamp(-5, 5)
            weights_aw = torch.exp(self.awac_beta * adv).clamp(max=100.0)
        ce = F.cross_entropy(
            logits.reshape(-1, self.vocab_size), actions.reshape(-1), reduction="none"
        ).reshape(B, -1)
        pi_loss = ops.masked_mean(weights_aw * ce, mask)

        loss = q_loss + v_loss + self.cql_weight * cql + pi_loss
        self.optimizer.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.parameters(), 1.0)
        self.optimizer.step()
        ops.polyak_update_(
            list(self.q1_target.parameters()) + list(self.q2_target.parameters()),
            list(self.q1_head.parameters()) + list(self.q2_head.parameters()),
            self.polyak,
        )
        return {
            "loss": float(loss.detach()),
            "q_loss": float(q_loss.detach()),
            "v_loss": float(v_loss.detach()),
            "cql": float(cql.detach()),
            "pi_loss": float(pi_loss.detach()),
        }

    # ------------------------------------------------------------------
    @torch.no_grad()
    def generate(
        self,
        idx: torch.Tensor,
        max_new_tokens: int,
        beta: float = 1.0,
        temperature: float = 1.0,
        top_k=None,
        top_p=None,
        eos_token_id=None,
        greedy: bool = False,
    ) -> torch.Tensor:
        """Decode with Q-V perturbed logits: ``logits/T + beta * (Q - V)``
        (reference ilql.py:933-1093 beta-perturbed sampling), with top-k /
        nucleus processing and EOS early stop."""
        from ..utils.sampling import process_logits

        B = idx.shape[0]
        done = torch.zeros(B, dtype=torch.bool, device=idx.device)
        for _ in range(max_new_tokens):
            ctx = idx[:, -self.gpt.max_positions :]
            hidden = self.gpt.transformer_forward(ctx)
            h_last = hidden[:, -1]
            logits = self.gpt.model["head"](h_last)
            q = torch.minimum(self.q1_head(h_last), self.q2_head(h_last))
            v = self.v_head(h_last)
            perturbed = logits / max(temperature, 1e-6) + beta * (q - v)
            perturbed = process_logits(perturbed, 1.0, top_k, top_p)
            if greedy:
                nxt = perturbed.argmax(-1, keepdim=True)
            else:
                nxt = torch.multinomial(F.softmax(perturbed, dim=-1), 1)
            if eos_token_id is not None:
                nxt = torch.where(done.unsqueeze(1), torch.full_like(nxt, eos_token_id), nxt)
                done = done | (nxt.squeeze(1) == eos_token_id)
            idx = torch.cat([idx, nxt], dim=1)
            if eos_token_id is not None and bool(done.all()):
                break
        return idx


class ILQL_Policy:
    """Acting interface over a token env (reference ilql.py:1335 /
    :2067 ``act``): history tokens in, beta-perturbed completion out."""

    def __init__(self, model: "ILQL", beta: float = 1.0, max_new_tokens: int = 32,
                 temperature: float = 1.0, top_k=None, top_p=None,
                 eos_token_id=None, greedy: bool = False):
        self.model = model
        self.beta = beta
        self.max_new_tokens = max_new_tokens
        self.temperature = temperature
        self.top_k = top_k
        self.top_p = top_p
        self.eos_token_id = eos_token_id
        self.greedy = greedy

    @torch.no_grad()
    def act(self, history_tokens: torch.Tensor) -> torch.Tensor:
        """(B, T) context -> (B, C) generated completion tokens."""
        if history_tokens.dim() == 1:
            history_tokens = history_tokens.unsqueeze(0)
        T = history_tokens.shape[1]
        full = self.model.generate(
            history_tokens.to(self.model.device), self.max_new_tokens,
            beta=self.beta, temperature=self.temperature,
            top_k=self.top_k, top_p=self.top_p,
            eos_token_id=self.eos_token_id, greedy=self.greedy,
        )
        return full[:, T:]


class ILQL_Evaluator:
    """Rollout scoring for ILQL policies (reference ilql.py:2089).

    ``env`` contract: ``reset() -> (B, T) prompt tokens``;
    ``score(full_sequences) -> (B,) rewards`` (same shape family as the
    LLM gyms).  Also reports the policy's value estimates on the prompts.
    """

    def __init__(self, env, n_batches: int = 1):
        self.env = env
        self.n_batches = n_batches

    @torch.no_grad()
    def evaluate(self, policy: ILQL_Policy) -> Dict[str, float]:
        import numpy as np

        rewards, values = [], []
        for _ in range(self.n_batches):
            prompts = self.env.reset()
            ids = prompts["input_ids"] if isinstance(prompts, dict) else prompts
            comp = policy.act(ids)
            full = torch.cat([ids.to(comp.device), comp], dim=1)
            r = self.env.score(full)
            rewards.extend(np.asarray(r, dtype=np.float32).tolist())
            hidden = policy.model.gpt.transformer_forward(
                ids.to(policy.model.device)
            )
            values.extend(
                policy.model.v_head(hidden[:, -1]).squeeze(-1).cpu().tolist()
            )
        return {
            "mean_reward": float(np.mean(rewards)),
            "std_reward": float(np.std(rewards)),
            "mean_value": float(np.mean(values)),
            "n": len(rewards),
        }
