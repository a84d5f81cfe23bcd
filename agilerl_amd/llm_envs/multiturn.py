"""Multi-turn LLM environments.

Reference parity: ``agilerl/llm_envs/token_observation.py:43``
(TokenObservationWrapper — tokenizes a text env into token space with
turn-boundary bookkeeping) and ``sync_vec_env.py:164``
(SyncMultiTurnVecEnv — batches B x group_size trajectories,
``get_trajectories`` :296 emits padded ids / action masks / turn ids /
rewards).  Re-shaped for this framework: the inner env speaks token
lists directly (text envs tokenize at the boundary), the vec env owns
trajectory accumulation and padding.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

import numpy as np
import torch

__all__ = ["MultiTurnTokenEnv", "TokenGuessEnv", "SyncMultiTurnVecEnv", "TextMultiTurnEnv", "SearchQAEnv"]


class MultiTurnTokenEnv:
    """One trajectory's logic, in token space.

    ``initial_prompt(rng) -> List[int]``
    ``respond(turn, completion_tokens, rng) -> (feedback_tokens, reward, done)``
    """

    def initial_prompt(self, rng: np.random.Generator) -> List[int]:
        raise NotImplementedError

    def respond(
        self, turn: int, completion: List[int], rng: np.random.Generator
    ) -> Tuple[List[int], float, bool]:
        raise NotImplementedError


class TokenGuessEnv(MultiTurnTokenEnv):
    """Synthetic 2-turn task: the first feedback reveals the target token;
    reward on the final turn is the fraction of completion tokens equal to
    the target.  A model that learns to copy feedback solves it — a
    learnable multi-turn credit-assignment signal for tests/benchmarks."""

    def __init__(self, vocab_size: int, prompt_len: int = 8, max_turns: int = 2):
        self.vocab_size = vocab_size
        self.prompt_len = prompt_len
        self.max_turns = max_turns
        self._target: Optional[int] = None

    def initial_prompt(self, rng: np.random.Generator) -> List[int]:
        self._target = int(rng.integers(1, self.vocab_size))
        return [int(x) for x in rng.integers(1, self.vocab_size, self.prompt_len)]

    def respond(self, turn: int, completion: List[int], rng: np.random.Generator):
        if turn + 1 >= self.max_turns:
            match = float(np.mean([t == self._target for t in completion])) if completion else 0.0
            return [], match, True
        # feedback: repeat the target token a few times as a strong hint
        return [self._target] * 4, 0.0, False


class SyncMultiTurnVecEnv:
    """Batches B prompts x group_size trajectories through turn-wise
    generation.  All trajectories advance together (sync); finished ones
    are frozen until the batch drains."""

    def __init__(
        self,
        env_factory: Callable[[], MultiTurnTokenEnv],
        data_batch_size: int = 2,
        group_size: int = 2,
        max_turns: int = 2,
        pad_token_id: int = 0,
        seed: Optional[int] = None,
    ):
        self.env_factory = env_factory
        self.data_batch_size = data_batch_size
        self.group_size = group_size
        self.max_turns = max_turns
        self.pad_token_id = pad_token_id
        self.rng = np.random.default_rng(seed)
        self.n_traj = data_batch_size * group_size
        self._envs: List[MultiTurnTokenEnv] = []
        # (tokens, is_action, turn, sampling_logps-or-None)
        self._segments: List[List[Tuple[List[int], bool, int, Optional[List[float]]]]] = []
        self._rewards: List[float] = []
        self._turn_rewards: List[List[float]] = []
        self._done: List[bool] = []
        self._turn = 0
        self.prompt_len = 0

    # ------------------------------------------------------------------
    def reset(self) -> Dict[str, torch.Tensor]:
        self._envs = []
        self._segments = []
        self._rewards = [0.0] * self.n_traj
        self._turn_rewards = [[0.0] * self.max_turns for _ in range(self.n_traj)]
        self._done = [False] * self.n_traj
        self._turn = 0
        for b in range(self.data_batch_size):
            env = self.env_factory()
            prompt = env.initial_prompt(self.rng)
            for _ in range(self.group_size):
                import copy as _copy

                env_g = _copy.deepcopy(env)
                self._envs.append(env_g)
                self._segments.append([(list(prompt), False, -1, None)])
        return self._current_prompts()

    def _history(self, i: int) -> List[int]:
        out: List[int] = []
        for tokens, _is_action, _turn, _lps in self._segments[i]:
            out.extend(tokens)
        return out

    def _current_prompts(self) -> Dict[str, torch.Tensor]:
        rows = [self._history(i) for i in range(self.n_traj)]
        P = max(len(r) for r in rows)
        self.prompt_len = P
        ids = torch.full((self.n_traj, P), self.pad_token_id, dtype=torch.long)
        am = torch.zeros((self.n_traj, P), dtype=torch.long)
        for i, r in enumerate(rows):  # left-pad for generation
            ids[i, P - len(r) :] = torch.tensor(r)
            am[i, P - len(r) :] = 1
        return {"input_ids": ids, "attention_mask": am}

    @property
    def all_done(self) -> bool:
        return all(self._done)

    # ------------------------------------------------------------------
    def step(
        self,
        sequences: torch.Tensor,
        sampling_logps: Optional[torch.Tensor] = None,
    ) -> Tuple[Optional[Dict[str, torch.Tensor]], bool]:
        """``sequences``: (n_traj, P + C) from generate on the last prompts;
        ``sampling_logps``: optional (n_traj, P+C-1) behavior-policy
        logprobs on the target grid (reference sync_vec_env.py:239 accepts
        captured vLLM sampling logprobs the same way)."""
        completions = sequences[:, self.prompt_len :].cpu()
        P = self.prompt_len
        for i in range(self.n_traj):
            if self._done[i]:
                continue
            comp = [int(t) for t in completions[i] if int(t) != self.pad_token_id]
            lps = None
            if sampling_logps is not None:
                lps = [float(x) for x in sampling_logps[i, P - 1 : P - 1 + len(comp)]]
            self._segments[i].append((comp, True, self._turn, lps))
            feedback, reward, done = self._envs[i].respond(self._turn, comp, self.rng)
            self._rewards[i] += float(reward)
            self._turn_rewards[i][self._turn] += float(reward)
            if feedback:
                self._segments[i].append((list(feedback), False, self._turn, None))
            self._done[i] = done
        self._turn += 1
        if self.all_done or self._turn >= self.max_turns:
            self._done = [True] * self.n_traj
            return None, True
        return self._current_prompts(), False

    # ------------------------------------------------------------------
    def get_trajectories(self) -> Dict[str, torch.Tensor]:
        """Right-padded full trajectories with completion-token action masks
        and per-target turn ids (-1 for non-action targets)."""
        rows, masks, turns, samp_rows = [], [], [], []
        for segs in self._segments:
            ids: List[int] = []
            act: List[float] = []
            trn: List[int] = []
            smp: List[float] = []
            for tokens, is_action, turn, lps in segs:
                ids.extend(tokens)
                act.extend([1.0 if is_action else 0.0] * len(tokens))
                trn.extend([turn] * len(tokens))
                if is_action and lps is not None and len(lps) == len(tokens):
                    smp.extend(lps)
                else:
                    smp.extend([0.0] * len(tokens))
            rows.append(ids)
            masks.append(act)
            turns.append(trn)
            samp_rows.append(smp)
        T = max(len(r) for r in rows)
        ids_t = torch.full((self.n_traj, T), self.pad_token_id, dtype=torch.long)
        am_t = torch.zeros((self.n_traj, T), dtype=torch.long)
        act_t = torch.zeros((self.n_traj, T), dtype=torch.float32)
        turn_t = torch.full((self.n_traj, T), -1, dtype=torch.long)
        samp_t = torch.zeros((self.n_traj, T), dtype=torch.float32)
        for i, (r, m, tr, sm) in enumerate(zip(rows, masks, turns, samp_rows)):
            ids_t[i, : len(r)] = torch.tensor(r)
            am_t[i, : len(r)] = 1
            act_t[i, : len(m)] = torch.tensor(m)
            turn_t[i, : len(tr)] = torch.tensor(tr)
            samp_t[i, : len(sm)] = torch.tensor(sm)
        # action_mask over TARGET positions j (predicting ids[:, j+1])
        action_mask = act_t[:, 1:]
        turn_ids = turn_t[:, 1:]
        return {
            "ids": ids_t,
            "attention_mask": am_t,
            "action_mask": action_mask,
            "turn_ids": turn_ids,
            "rewards": torch.tensor(self._rewards, dtype=torch.float32),
            "turn_rewards": torch.tensor(self._turn_rewards, dtype=torch.float32),
            "sampling_logps": samp_t[:, 1:],
        }


class TextMultiTurnEnv(MultiTurnTokenEnv):
    """Tokenizer boundary for text-level multi-turn tasks.

    Reference parity: ``agilerl/llm_envs/token_observation.py:43``
    (TokenObservationWrapper) — the conversation lives as text turns; the
    tokenizer converts at the env boundary so the vec env / algorithms
    stay purely token-space.  Subclasses implement ``initial_message``
    and ``respond_text``; feedback is rendered as a new user turn through
    the chat template (or a plain role-prefixed format without one).
    """

    def __init__(self, tokenizer, system_prompt: Optional[str] = None,
                 max_prompt_tokens: int = 512):
        self.tokenizer = tokenizer
        self.system_prompt = system_prompt
        self.max_prompt_tokens = max_prompt_tokens

    # -- subclass hooks -------------------------------------------------
    def initial_message(self, rng: np.random.Generator) -> str:
        raise NotImplementedError

    def respond_text(
        self, turn: int, completion_text: str, rng: np.random.Generator
    ) -> Tuple[str, float, bool]:
        """Returns (feedback_text, reward, done); feedback ignored when done."""
        raise NotImplementedError

    # -- token boundary -------------------------------------------------
    def _encode(self, text: str) -> List[int]:
        ids = self.tokenizer.encode(text)
        return list(ids[-self.max_prompt_tokens:])

    def initial_prompt(self, rng: np.random.Generator) -> List[int]:
        from ..llm.chat import apply_chat_template

        text = apply_chat_template(
            self.tokenizer, self.initial_message(rng), system_prompt=self.system_prompt
        )
        return self._encode(text)

    def respond(self, turn: int, completion: List[int], rng: np.random.Generator):
        completion_text = self.tokenizer.decode(completion)
        feedback, reward, done = self.respond_text(turn, completion_text, rng)
        if done:
            return [], reward, True
        return self._encode(f"\nUser: {feedback}\nAssistant:"), reward, False


class SearchQAEnv(TextMultiTurnEnv):
    """Tool-use QA over an offline corpus (reference search.py flow):
    turn 0 the model may emit ``<tool>query</tool>`` calls; feedback is
    the retrieved documents; the final turn is scored by
    ``<answer>...</answer>`` match against the gold answer, with a small
    format bonus for issuing a well-formed tool call."""

    def __init__(self, tokenizer, documents, questions, answers,
                 max_turns: int = 2, tool_k: int = 2, **kw):
        super().__init__(tokenizer, **kw)
        from .search import SearchTool

        self.tool = SearchTool(documents)
        self.questions = list(questions)
        self.answers = list(answers)
        self.max_turns = max_turns
        self.tool_k = tool_k
        self._answer = None

    def initial_message(self, rng: np.random.Generator) -> str:
        i = int(rng.integers(0, len(self.questions)))
        self._answer = str(self.answers[i])
        return (
            f"{self.questions[i]}\n"
            "Use <tool>query</tool> to search, then answer with <answer>...</answer>."
        )

    def respond_text(self, turn, completion_text, rng):
        from .search import extract_answer, parse_tool_calls

        final = turn + 1 >= self.max_turns
        answer = extract_answer(completion_text)
        if answer is not None or final:
            correct = answer is not None and self._answer.lower() in answer.lower()
            return "", (1.0 if correct else 0.0), True
        calls = parse_tool_calls(completion_text)
        if not calls:
            return "No tool call found. Answer with <answer>...</answer>.", 0.0, False
        docs = self.tool(calls[0], k=self.tool_k)
        joined = "\n".join(docs) if docs else "(no results)"
        return f"Search results:\n{joined}", 0.05, False
