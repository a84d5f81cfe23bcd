"""HTTP agent-serving tests (FastAPI TestClient, no real sockets)."""

import numpy as np
import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from agilerl_amd.algorithms import DQN, PPO
from agilerl_amd.serve import create_app, load_agent
from agilerl_amd.spaces import Box, Discrete

NET = {"arch": "mlp", "hidden_size": [16]}


@pytest.fixture
def dqn_checkpoint(tmp_path):
    agent = DQN(Box(-1.0, 1.0, (4,)), Discrete(2), net_config=dict(NET))
    agent.fitness.append(123.0)
    path = str(tmp_path / "dqn.pt")
    agent.save_checkpoint(path)
    return path


def test_serve_roundtrip(dqn_checkpoint):
    agent = load_agent(dqn_checkpoint)
    client = TestClient(create_app(agent))
    assert client.get("/healthz").json()["status"] == "ok"
    info = client.get("/info").json()
    assert info["algo"] == "DQN"
    assert info["action_space"]["n"] == 2
    assert info["fitness_tail"][-1] == 123.0

    r = client.post("/predict", json={"obs": np.random.randn(3, 4).tolist()})
    assert r.status_code == 200
    actions = r.json()["action"]
    assert len(actions) == 3 and all(a in (0, 1) for a in actions)

    # single-observation mode
    r = client.post("/predict", json={"obs": [0.1, 0.2, 0.3, 0.4], "batch": False})
    assert r.json()["action"] in (0, 1)

    # bad shape -> 400, not 500
    r = client.post("/predict", json={"obs": [[1.0, 2.0]]})
    assert r.status_code == 400


def test_serve_reload(dqn_checkpoint, tmp_path):
    agent = load_agent(dqn_checkpoint)
    client = TestClient(create_app(agent))
    ppo = PPO(Box(-1.0, 1.0, (4,)), Discrete(2), net_config=dict(NET))
    ppo_path = str(tmp_path / "ppo.pt")
    ppo.save_checkpoint(ppo_path)
    r = client.post("/reload", json={"path": ppo_path})
    assert r.json() == {"status": "reloaded", "algo": "PPO"}
    assert client.get("/info").json()["algo"] == "PPO"
    r = client.post("/reload", json={"path": str(tmp_path / "missing.pt")})
    assert r.status_code == 400


def test_serve_llm_generate():
    import torch

    from agilerl_amd.algorithms.llm.grpo import GRPO

    tiny = dict(model_type="llama", vocab_size=64, hidden_size=32, intermediate_size=64,
                num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=1,
                max_position_embeddings=64, pad_token_id=0)
    agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                 max_completion_tokens=4)
    client = TestClient(create_app(agent))
    r = client.post("/generate", json={"input_ids": [1, 2, 3], "max_new_tokens": 4})
    assert r.status_code == 200
    body = r.json()
    assert len(body["output_ids"][0]) == 7
    assert len(body["completion_ids"][0]) == 4
    # classic agents refuse cleanly
    r = client.post("/generate", json={"prompt": "hi"})
    assert r.status_code in (200, 400)  # GRPO without tokenizer -> 400


def test_serve_llm_generate_with_messages():
    import torch

    from agilerl_amd.algorithms.llm.grpo import GRPO

    class WordTok:
        chat_template = None
        pad_token_id = 0

        def encode(self, text):
            return [(hash(w) % 60) + 1 for w in text.split()][:16]

        def decode(self, ids):
            return " ".join(f"t{int(i)}" for i in ids)

    tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                intermediate_size=64, num_hidden_layers=1,
                num_attention_heads=2, num_key_value_heads=1,
                max_position_embeddings=128, pad_token_id=0)
    agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                 max_completion_tokens=4, tokenizer=WordTok())
    client = TestClient(create_app(agent))
    r = client.post("/generate", json={
        "messages": [{"role": "user", "content": "hello there"}],
        "max_new_tokens": 3,
    })
    assert r.status_code == 200, r.text
    body = r.json()
    assert len(body["completion_ids"][0]) == 3
    assert isinstance(body["completion"][0], str)


def test_serve_generate_uses_paged_engine_when_configured():
    import torch

    from agilerl_amd.algorithms.llm.grpo import GRPO

    tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                intermediate_size=64, num_hidden_layers=1,
                num_attention_heads=2, num_key_value_heads=1,
                max_position_embeddings=128, pad_token_id=0)
    agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                 max_completion_tokens=4, generation="paged")
    client = TestClient(create_app(agent))
    r = client.post("/generate", json={"input_ids": [1, 2, 3], "max_new_tokens": 3,
                                       "do_sample": False})
    assert r.status_code == 200
    assert len(r.json()["completion_ids"][0]) == 3
    assert agent._decode_engine is not None  # the paged engine actually ran


class TestStreamingGenerate:
    def test_stream_generate_tokens_match_paged(self):
        import torch
        """stream_generate yields exactly the tokens generate_paged
        produces for the same greedy prompt."""
        from agilerl_amd.algorithms.llm.grpo import GRPO

        tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                    intermediate_size=64, num_hidden_layers=1,
                    num_attention_heads=2, num_key_value_heads=1,
                    max_position_embeddings=128, pad_token_id=0)
        torch.manual_seed(0)
        agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                     max_completion_tokens=6, generation="paged")
        torch.manual_seed(1)
        ids = torch.randint(1, 64, (1, 5))
        mask = torch.ones_like(ids)
        full = agent.generate_paged(ids, mask, do_sample=False)
        want = full[0, 5:].tolist()
        agent._decode_engine = None  # independent engine for the stream
        got = list(agent.stream_generate(ids[0], mask[0], do_sample=False))
        assert got == want

    def test_http_stream_endpoint_ndjson(self):
        import json

        import torch

        from starlette.testclient import TestClient

        from agilerl_amd.algorithms.llm.grpo import GRPO
        from agilerl_amd.serve import create_app

        tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                    intermediate_size=64, num_hidden_layers=1,
                    num_attention_heads=2, num_key_value_heads=1,
                    max_position_embeddings=128, pad_token_id=0)
        torch.manual_seed(0)
        agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                     max_completion_tokens=5, generation="paged")
        client = TestClient(create_app(agent))
        with client.stream("POST", "/generate/stream", json={
            "input_ids": [3, 7, 11], "max_new_tokens": 5, "do_sample": False,
        }) as r:
            assert r.status_code == 200
            lines = [json.loads(l) for l in r.iter_lines() if l.strip()]
        assert lines[-1]["done"] is True
        token_events = [l for l in lines[:-1] if "token_id" in l]
        assert len(token_events) == 5
        assert lines[-1]["completion_ids"] == [e["token_id"] for e in token_events]
