"""Error contracts: invalid configurations fail fast with clear messages."""

import pytest
import torch

from agilerl_amd.spaces import Box

NET = {"arch": "mlp", "hidden_size": [16]}


def test_grpo_rejects_bad_generation_mode():
    from agilerl_amd.algorithms.llm.grpo import GRPO

    tiny = dict(model_type="llama", vocab_size=32, hidden_size=16,
                intermediate_size=32, num_hidden_layers=1,
                num_attention_heads=2, num_key_value_heads=1,
                max_position_embeddings=32, pad_token_id=0)
    with pytest.raises(ValueError, match="generation"):
        GRPO(model_config=tiny, dtype=torch.float32, generation="vllm")


def test_unknown_env_id_lists_known():
    from agilerl_amd.envs import make_vect_envs

    with pytest.raises(KeyError, match="CartPole-v1"):
        make_vect_envs("NoSuchEnv-v0", num_envs=2)


def test_group_advantage_rejects_indivisible_batch():
    from agilerl_amd import ops

    with pytest.raises(RuntimeError):
        ops.group_advantage(torch.randn(5), group_size=2)


def test_manifest_unknown_algo():
    from agilerl_amd.models.manifest import TrainingManifest, resolve_algo_class

    with pytest.raises(KeyError, match="Registered"):
        TrainingManifest.model_validate(
            {"algorithm": {"name": "NotAnAlgo"},
             "environment": {"env_id": "CartPole-v1"}}
        )
    with pytest.raises(KeyError, match="Registered"):
        resolve_algo_class("NotAnAlgo")


def test_paged_cache_double_alloc_and_unknown_free():
    from agilerl_amd.llm.paged_cache import PagedKVCache

    c = PagedKVCache(1, 1, 4, num_pages=2, page_size=2)
    c.alloc(0)
    with pytest.raises(KeyError):
        c.alloc(0)
    c.free(123)  # unknown id is a no-op, not a crash


def test_dqn_rejects_continuous_space():
    from agilerl_amd.algorithms import DQN

    with pytest.raises(TypeError, match="discrete action space"):
        DQN(Box(-1.0, 1.0, (4,)), Box(-1.0, 1.0, (2,)), net_config=dict(NET))
