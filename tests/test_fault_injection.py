"""Failure detection / recovery: injected faults must
surface loudly and resume must recover from the last checkpoint."""

import os

import numpy as np
import pytest
import torch


class _FaultyEnv:
    """CartPole-shaped env that raises after N steps (worker crash model)."""

    def __init__(self, wrapped, fail_after: int):
        self._env = wrapped
        self._steps = 0
        self._fail_after = fail_after

    def __getattr__(self, name):
        return getattr(self._env, name)

    def reset(self, *a, **k):
        return self._env.reset(*a, **k)

    def step(self, actions):
        self._steps += 1
        if self._steps >= self._fail_after:
            raise RuntimeError("injected env fault")
        return self._env.step(actions)


class TestFaultInjection:
    def test_env_fault_surfaces_from_training_loop(self):
        from agilerl_amd.algorithms import DQN
        from agilerl_amd.components import ReplayBuffer
        from agilerl_amd.envs import CartPoleVecEnv
        from agilerl_amd.training.train_off_policy import train_off_policy

        env = _FaultyEnv(CartPoleVecEnv(4, seed=0), fail_after=20)
        pop = [DQN(env.single_observation_space, env.single_action_space)]
        memory = ReplayBuffer(1000)
        with pytest.raises(RuntimeError, match="injected env fault"):
            train_off_policy(
                env, "CartPole-v1", "DQN", pop, memory,
                max_steps=10_000, evo_steps=500, verbose=False,
            )

    def test_checkpoint_resume_after_crash(self, tmp_path):
        """Crash mid-training -> the checkpoint written before the crash
        restores an agent with the same weights and training counters."""
        from agilerl_amd.algorithms import DQN
        from agilerl_amd.algorithms.core.base import EvolvableAlgorithm
        from agilerl_amd.spaces import Box, Discrete

        torch.manual_seed(3)
        agent = DQN(Box(-1, 1, (4,)), Discrete(2))
        batch = {
            "obs": torch.randn(16, 4), "action": torch.randint(0, 2, (16, 1)),
            "reward": torch.randn(16, 1), "next_obs": torch.randn(16, 4),
            "done": torch.zeros(16, 1),
        }
        for _ in range(3):
            agent.learn(batch)
        agent.steps = [777]
        agent.fitness = [42.0]
        ckpt = tmp_path / "pre_crash.pt"
        agent.save_checkpoint(str(ckpt))
        with torch.no_grad():  # "crash" corrupts the live agent
            for p in agent.actor.parameters():
                p.fill_(float("nan"))

        restored = EvolvableAlgorithm.load(str(ckpt))
        x = torch.randn(5, 4)
        assert torch.isfinite(restored.actor(x)).all()
        assert restored.steps == [777]
        assert restored.fitness == [42.0]
        # optimizer moments survived too: one more learn moves weights
        before = [p.detach().clone() for p in restored.actor.parameters()]
        restored.learn(batch)
        assert any(
            not torch.equal(a, b)
            for a, b in zip(before, restored.actor.parameters())
        )

    def test_max_wall_seconds_budget_stop(self):
        """wall-clock budget stops the multiturn loop."""
        from agilerl_amd.algorithms.llm.grpo import GRPO
        from agilerl_amd.llm_envs.multiturn import SyncMultiTurnVecEnv, TokenGuessEnv
        from agilerl_amd.training.llm.multiturn import finetune_llm_multiturn

        tiny = dict(model_type="llama", vocab_size=32, hidden_size=32,
                    intermediate_size=64, num_hidden_layers=1,
                    num_attention_heads=2, num_key_value_heads=1,
                    max_position_embeddings=128, pad_token_id=0)
        agent = GRPO(model_config=tiny, dtype=torch.float32,
                     lora_config={"r": 2}, group_size=2, micro_batch_size=2,
                     max_completion_tokens=4)
        env = SyncMultiTurnVecEnv(lambda: TokenGuessEnv(vocab_size=32),
                                  data_batch_size=2, group_size=2, max_turns=2)
        import time

        t0 = time.time()
        finetune_llm_multiturn(
            env, [agent], max_steps=10_000, evo_steps=2,
            max_wall_seconds=2.0, verbose=False,
        )
        assert time.time() - t0 < 60  # stopped by budget, not max_steps
