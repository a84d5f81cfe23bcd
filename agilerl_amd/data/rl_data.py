"""Token-level RL sample construction for the offline (ILQL/BC) stack.

Reference parity: ``agilerl/data/rl_data.py`` — TokenReward shaping
(:17-52), DataPoint (:53: dialogue -> tokens + state/action index lists +
per-token rewards with the utterance reward folded into each action's
last token), RL_Dataset list/iterable variants (:175-288).

MI355X note: ``DataPoint.to_tensors``/``collate`` emit padded batch
tensors shaped for :class:`agilerl_amd.algorithms.ilql.ILQL`'s
``learn`` contract (ids / per-target rewards / target mask), so a
dataset of dialogues feeds the HIP-kernel training path directly.
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import torch

from .language_environment import LanguageObservation
from .tokenizer import DialogueTokenizer

__all__ = ["TokenReward", "ConstantTokenReward", "SpecifiedTokenReward",
           "DataPoint", "ListRLDataset", "IterableRLDataset"]


class TokenReward(ABC):
    """Per-token shaping reward added on top of utterance rewards."""

    @abstractmethod
    def get_token_reward(self, tokens: List[int]) -> List[float]: ...


class ConstantTokenReward(TokenReward):
    def __init__(self, c: float = 0.0):
        self.c = float(c)

    def get_token_reward(self, tokens: List[int]) -> List[float]:
        return [self.c] * len(tokens)


class SpecifiedTokenReward(TokenReward):
    """Lookup table token -> reward (default ``scale * default``)."""

    def __init__(self, token_data: Dict[int, float], scale: float = 1.0,
                 shift: float = 0.0):
        self.token_data = dict(token_data)
        self.scale = float(scale)
        self.shift = float(shift)

    def get_token_reward(self, tokens: List[int]) -> List[float]:
        return [self.token_data.get(int(t), 0.0) * self.scale + self.shift
                for t in tokens]


@dataclass
class DataPoint:
    raw_str: str
    tokens: List[int]
    state_idxs: List[int]
    action_idxs: List[int]
    rewards: List[float]          # per action token
    terminals: List[int]
    utterance_state_idxs: List[int]
    utterance_action_idxs: List[int]
    utterance_rewards: List[float]
    utterance_terminals: List[int]
    meta: Optional[Dict[str, Any]] = field(default=None)

    @classmethod
    def from_obs(cls, obs: LanguageObservation, tokenizer: DialogueTokenizer,
                 token_reward: Optional[TokenReward] = None,
                 meta: Optional[Dict[str, Any]] = None) -> "DataPoint":
        token_reward = token_reward or ConstantTokenReward(0.0)
        sequence, terminal = obs.to_sequence()
        obs_meta = obs.metadata()
        if obs_meta:
            meta = {**obs_meta, **(meta or {})}

        # dialogue string with boundary markers: env turns end with <eos>,
        # agent turns with <eoa>; opener marks who speaks first
        first_is_agent = bool(sequence) and sequence[0][1] is not None
        raw = tokenizer.id_to_token(
            tokenizer.boa_token_id if first_is_agent else tokenizer.bos_token_id
        )
        action_rewards: List[float] = []
        for utterance, reward in sequence:
            raw += utterance
            if reward is None:
                raw += tokenizer.id_to_token(tokenizer.eos_token_id)
            else:
                raw += tokenizer.id_to_token(tokenizer.eoa_token_id)
                action_rewards.append(float(reward))
        if terminal:
            raw += tokenizer.id_to_token(tokenizer.eod_token_id)

        tokens = tokenizer.encode(raw)
        tok_rewards = token_reward.get_token_reward(tokens)

        state_idxs: List[int] = []
        action_idxs: List[int] = []
        rewards: List[float] = []
        u_state, u_action, u_rewards = [], [], []
        span_start, action_i = 0, 0
        for i, t in enumerate(tokens):
            if t == tokenizer.eos_token_id:
                span_start = i
            elif t == tokenizer.eoa_token_id:
                # agent span (span_start, i): every token is an action
                idxs = list(range(span_start, i))
                action_idxs.extend(idxs)
                state_idxs.extend(idxs)
                span_rewards = [tok_rewards[x] for x in idxs]
                if span_rewards:
                    span_rewards[-1] += action_rewards[action_i]
                rewards.extend(span_rewards)
                u_action.append(i)
                u_state.append(span_start)
                u_rewards.append(action_rewards[action_i]
                                 + sum(tok_rewards[x] for x in idxs))
                span_start = i
                action_i += 1
        state_idxs.append(len(tokens) - 1)
        u_state.append(len(tokens) - 1)
        terminals = [0] * (len(state_idxs) - 1) + [int(terminal)]
        u_terminals = [0] * (len(u_state) - 1) + [int(terminal)]
        return cls(raw, tokens, state_idxs, action_idxs, rewards, terminals,
                   u_state, u_action, u_rewards, u_terminals, meta=meta)

    def to_tensors(self, device="cpu", max_length: Optional[int] = None):
        tok = torch.tensor(self.tokens, device=device)
        s = torch.tensor(self.state_idxs, dtype=torch.long, device=device)
        a = torch.tensor(self.action_idxs, dtype=torch.long, device=device)
        r = torch.tensor(self.rewards, device=device)
        term = torch.tensor(self.terminals, device=device)
        if max_length is not None:
            tok = tok[:max_length]
            s = s[s < max_length]
            keep = a < max_length - 1
            a, r = a[keep], r[keep]
            term = term[: s.shape[0]]
        return tok, s, a, r, term

    def to_ilql_batch_row(self, max_length: int) -> Dict[str, torch.Tensor]:
        """One padded ILQL row: ids (T,), per-target rewards (T-1,),
        target mask (T-1,) marking agent-action targets."""
        T = int(max_length)
        ids = torch.zeros(T, dtype=torch.long)
        n = min(len(self.tokens), T)
        ids[:n] = torch.tensor(self.tokens[:n])
        rewards = torch.zeros(T - 1)
        mask = torch.zeros(T - 1)
        for idx, rew in zip(self.action_idxs, self.rewards):
            # action_idxs are hidden-state positions: position idx predicts
            # the action token at idx+1, i.e. target slot idx in the
            # (T-1,)-shaped ILQL layout
            if 0 <= idx < T - 1:
                mask[idx] = 1.0
                rewards[idx] = rew
        return {"ids": ids, "rewards": rewards, "mask": mask}

    @staticmethod
    def collate(points: List["DataPoint"], max_length: Optional[int] = None,
                device="cpu") -> Dict[str, torch.Tensor]:
        T = max_length or max(len(p.tokens) for p in points)
        rows = [p.to_ilql_batch_row(T) for p in points]
        return {k: torch.stack([r[k] for r in rows]).to(device)
                for k in rows[0]}


class _RLDatasetBase:
    def __init__(self, tokenizer: DialogueTokenizer,
                 token_reward: Optional[TokenReward] = None,
                 max_len: Optional[int] = None):
        self.tokenizer = tokenizer
        self.token_reward = token_reward or ConstantTokenReward(0.0)
        self.max_len = max_len

    def datapoint(self, obs: LanguageObservation) -> DataPoint:
        return DataPoint.from_obs(obs, self.tokenizer, self.token_reward)


class ListRLDataset(_RLDatasetBase):
    """Finite dataset over a list of observations (reference
    rl_data.py:270 List_RL_Dataset)."""

    def __init__(self, observations: List[LanguageObservation],
                 tokenizer: DialogueTokenizer,
                 token_reward: Optional[TokenReward] = None,
                 max_len: Optional[int] = None):
        super().__init__(tokenizer, token_reward, max_len)
        self.observations = list(observations)

    def size(self) -> int:
        return len(self.observations)

    __len__ = size

    def get_item(self, idx: int) -> DataPoint:
        return self.datapoint(self.observations[idx])

    __getitem__ = get_item

    def sample_batch(self, batch_size: int, device="cpu") -> Dict[str, torch.Tensor]:
        import numpy as np

        idx = np.random.randint(0, len(self), size=batch_size)
        points = [self.get_item(int(i)) for i in idx]
        return DataPoint.collate(points, self.max_len, device)


class IterableRLDataset(_RLDatasetBase):
    """Streaming dataset over an observation generator (reference
    rl_data.py:282 Iterable_RL_Dataset)."""

    def __init__(self, observation_iter, tokenizer: DialogueTokenizer,
                 token_reward: Optional[TokenReward] = None,
                 max_len: Optional[int] = None):
        super().__init__(tokenizer, token_reward, max_len)
        self._iter = iter(observation_iter)

    def sample_item(self) -> DataPoint:
        return self.datapoint(next(self._iter))

    def sample_batch(self, batch_size: int, device="cpu") -> Dict[str, torch.Tensor]:
        points = [self.sample_item() for _ in range(batch_size)]
        return DataPoint.collate(points, self.max_len, device)
