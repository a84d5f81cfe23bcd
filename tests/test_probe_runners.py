"""check_*_with_probe_env runner API (reference probe_envs check_* parity)."""

import numpy as np
import pytest
import torch

from agilerl_amd.algorithms import DQN, PPO
from agilerl_amd.envs.probe import (
    ConstantRewardEnv,
    FixedObsPolicyEnv,
    check_llm_policy_with_probe_env,
    check_on_policy_with_probe_env,
    check_q_learning_with_probe_env,
)

NET = {"arch": "mlp", "hidden_size": [32]}


def test_q_learning_runner():
    np.random.seed(0), torch.manual_seed(0)
    agent = check_q_learning_with_probe_env(
        ConstantRewardEnv(num_envs=4), DQN,
        dict(lr=1e-2, tau=0.1, batch_size=64, net_config=dict(NET)),
    )
    assert agent is not None


def test_on_policy_runner():
    np.random.seed(0), torch.manual_seed(0)
    check_on_policy_with_probe_env(
        FixedObsPolicyEnv(num_envs=8), PPO,
        dict(lr=5e-3, batch_size=64, ent_coef=0.0, net_config=dict(NET)),
    )


@pytest.mark.slow
def test_grpo_constant_target_reward_improves():
    """Analytic LLM probe (reference probe_envs_llm.py analog): reward is the
    fraction of completion tokens equal to a fixed target token, so the
    optimal policy is a unigram shift GRPO must discover from group-relative
    advantages.  Chance level is 1/32; training must clearly beat it."""
    np.random.seed(0), torch.manual_seed(0)
    from agilerl_amd.algorithms.llm.grpo import GRPO
    from agilerl_amd.llm_envs import TokenReasoningGym

    TARGET = 7

    def reward_fn(seqs, prompt_len):
        comp = seqs[:, prompt_len:]
        return (comp == TARGET).float().mean(dim=1).cpu().numpy()

    model_cfg = dict(
        model_type="llama", vocab_size=32, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, pad_token_id=0,
    )
    agent = GRPO(model_config=model_cfg, dtype=torch.float32,
                 lora_config={"r": 8, "lora_alpha": 16}, micro_batch_size=16,
                 group_size=8, lr=1e-2, beta=0.0, max_completion_tokens=4)
    env = TokenReasoningGym(vocab_size=32, prompt_len=4, data_batch_size=2,
                            group_size=8, reward_fn=reward_fn, seed=0)
    hist = check_llm_policy_with_probe_env(agent, env, iterations=25)
    assert np.mean(hist[-8:]) > np.mean(hist[:3]) + 0.02, hist


def test_continuous_q_learning_runner_ddpg():
    from agilerl_amd.algorithms import DDPG
    from agilerl_amd.envs.probe import (
        ConstantRewardContActionsEnv,
        check_policy_q_learning_with_probe_env,
    )

    np.random.seed(0), torch.manual_seed(0)
    agent = check_policy_q_learning_with_probe_env(
        ConstantRewardContActionsEnv(num_envs=4), DDPG,
        dict(lr_actor=1e-3, lr_critic=1e-2, tau=0.1, batch_size=64,
             net_config=dict(NET)),
    )
    assert agent is not None


def test_continuous_q_learning_runner_td3():
    from agilerl_amd.algorithms import TD3
    from agilerl_amd.envs.probe import (
        ConstantRewardContActionsEnv,
        check_policy_q_learning_with_probe_env,
    )

    np.random.seed(1), torch.manual_seed(1)
    agent = check_policy_q_learning_with_probe_env(
        ConstantRewardContActionsEnv(num_envs=4), TD3,
        dict(lr_actor=1e-3, lr_critic=1e-2, tau=0.1, batch_size=64,
             net_config=dict(NET)),
    )
    assert agent is not None
