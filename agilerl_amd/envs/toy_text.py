"""Batched toy-text envs (gymnasium toy_text family, first-party).

``BlackjackVecEnv`` mirrors Gymnasium Blackjack-v1: infinite deck,
observation = (player sum, dealer showing card, usable ace) as a Tuple
of Discretes — the reference's ``multi_input.yaml`` config uses it to
exercise the Tuple-observation multi-input encoder.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np

from ..spaces import Discrete, TupleSpace

__all__ = ["BlackjackVecEnv"]


def _draw(rng, n) -> np.ndarray:
    # infinite deck: 1-9 uniform, 10 with weight 4 (10/J/Q/K)
    return np.minimum(rng.integers(1, 14, size=n), 10)


class BlackjackVecEnv:
    """N parallel Blackjack hands; actions: 0 = stick, 1 = hit."""

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None,
                 natural: bool = False, sab: bool = False):
        self.num_envs = int(num_envs)
        self.rng = np.random.default_rng(seed)
        self.natural = bool(natural)
        self.sab = bool(sab)
        self.single_observation_space = TupleSpace(
            (Discrete(32), Discrete(11), Discrete(2))
        )
        self.single_action_space = Discrete(2)
        self.observation_space = self.single_observation_space
        self.action_space = self.single_action_space
        N = self.num_envs
        self.player_sum = np.zeros(N, dtype=np.int64)
        self.usable_ace = np.zeros(N, dtype=np.int64)
        self.dealer_show = np.zeros(N, dtype=np.int64)
        self.dealer_hole = np.zeros(N, dtype=np.int64)

    # ------------------------------------------------------------------
    def _deal_rows(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        if n == 0:
            return
        c1, c2 = _draw(self.rng, n), _draw(self.rng, n)
        total = c1 + c2
        ace = (c1 == 1) | (c2 == 1)
        total = np.where(ace & (total + 10 <= 21), total + 10, total)
        self.player_sum[mask] = total
        self.usable_ace[mask] = (ace & (total <= 21)).astype(np.int64)
        self.dealer_show[mask] = _draw(self.rng, n)
        self.dealer_hole[mask] = _draw(self.rng, n)

    def _obs(self) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
        return (self.player_sum.copy(), self.dealer_show.copy(),
                self.usable_ace.copy())

    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self.rng = np.random.default_rng(seed)
        self._deal_rows(np.ones(self.num_envs, dtype=bool))
        return self._obs(), {}

    def _dealer_total(self, idx: np.ndarray) -> np.ndarray:
        """Play the dealer out (hit to 17) for the selected rows."""
        total = self.dealer_show[idx] + self.dealer_hole[idx]
        ace = (self.dealer_show[idx] == 1) | (self.dealer_hole[idx] == 1)
        soft = ace & (total + 10 <= 21)
        total = np.where(soft, total + 10, total)
        active = total < 17
        while active.any():
            card = _draw(self.rng, int(active.sum()))
            t = total[active]
            s = soft[active]
            a = ace[active] | (card == 1)
            t = t + card
            # demote a soft ace on bust; promote a new ace when it fits
            demote = s & (t > 21)
            t = np.where(demote, t - 10, t)
            s = np.where(demote, False, s)
            promote = a & ~s & (t + 10 <= 21)
            t = np.where(promote, t + 10, t)
            s = s | promote
            total[active], soft[active], ace[active] = t, s, a
            active = total < 17
        return total

    def step(self, actions):
        actions = np.asarray(actions).reshape(-1).astype(np.int64)
        N = self.num_envs
        reward = np.zeros(N, dtype=np.float32)
        term = np.zeros(N, dtype=bool)

        hit = actions == 1
        if hit.any():
            card = _draw(self.rng, int(hit.sum()))
            t = self.player_sum[hit] + card
            ace = self.usable_ace[hit].astype(bool) | (card == 1)
            soft = self.usable_ace[hit].astype(bool)
            demote = soft & (t > 21)
            t = np.where(demote, t - 10, t)
            soft = np.where(demote, False, soft)
            promote = ace & ~soft & (t + 10 <= 21)
            t = np.where(promote, t + 10, t)
            soft = soft | promote
            self.player_sum[hit] = t
            self.usable_ace[hit] = soft.astype(np.int64)
            bust = np.zeros(N, dtype=bool)
            bust[hit] = t > 21
            reward[bust] = -1.0
            term |= bust

        stick = (actions == 0) & ~term
        if stick.any():
            dealer = self._dealer_total(stick)
            player = self.player_sum[stick]
            r = np.where(dealer > 21, 1.0,
                         np.sign(player - dealer).astype(np.float32))
            reward[stick] = r
            term |= stick

        info = {}
        obs = self._obs()
        if term.any():
            info["final_observation"] = tuple(o.copy() for o in obs)
            self._deal_rows(term)
            obs = self._obs()
        return obs, reward, term, np.zeros(N, dtype=bool), info

    def close(self) -> None:
        pass
