"""EvolvableGPT / EvolvableBERT / ILQL / BC_LM tests."""

import numpy as np
import torch

from agilerl_amd.algorithms.bc_lm import BC_LM
from agilerl_amd.algorithms.ilql import ILQL
from agilerl_amd.modules.bert import EvolvableBERT
from agilerl_amd.modules.gpt import EvolvableGPT


class TestEvolvableGPT:
    def test_forward_and_loss(self):
        gpt = EvolvableGPT(vocab_size=64, n_layer=2, n_head=4, n_embd=32, max_positions=32)
        ids = torch.randint(0, 64, (3, 16))
        logits = gpt(ids)
        assert logits.shape == (3, 16, 64)
        logits, loss = gpt(ids[:, :-1], ids[:, 1:])
        assert torch.isfinite(loss)

    def test_generate(self):
        gpt = EvolvableGPT(vocab_size=64, n_layer=2, n_head=4, n_embd=32, max_positions=32)
        out = gpt.generate(torch.randint(0, 64, (2, 4)), max_new_tokens=5)
        assert out.shape == (2, 9)

    def test_mutations_preserve(self):
        gpt = EvolvableGPT(vocab_size=64, n_layer=2, n_head=4, n_embd=32, max_positions=32)
        w = gpt.model["wte"].weight.detach().clone()
        gpt.add_layer()
        assert gpt.n_layer == 3
        assert torch.equal(gpt.model["wte"].weight[:, :32], w)
        gpt.add_node(numb_new_nodes=32)
        assert gpt.n_embd == 64
        assert gpt.n_embd % gpt.n_head == 0
        ids = torch.randint(0, 64, (2, 8))
        assert gpt(ids).shape == (2, 8, 64)

    def test_clone(self):
        gpt = EvolvableGPT(vocab_size=64, n_layer=2, n_head=4, n_embd=32, max_positions=32)
        gpt.add_layer()
        clone = gpt.clone()
        ids = torch.randint(0, 64, (2, 8))
        gpt.eval(), clone.eval()
        torch.testing.assert_close(gpt(ids), clone(ids))


class TestEvolvableBERT:
    def test_pooled_output(self):
        bert = EvolvableBERT(vocab_size=64, n_layer=2, n_head=4, n_embd=32,
                             max_positions=32, num_outputs=5)
        ids = torch.randint(0, 64, (3, 16))
        out = bert(ids)
        assert out.shape == (3, 5)

    def test_not_causal(self):
        bert = EvolvableBERT(vocab_size=64, n_layer=1, n_head=2, n_embd=16,
                             max_positions=16, num_outputs=2)
        bert.eval()
        # flipping a LATE token must change the pooled (first-token) output
        ids = torch.randint(0, 64, (1, 8))
        ids2 = ids.clone()
        ids2[0, -1] = (ids2[0, -1] + 1) % 64
        assert not torch.allclose(bert(ids), bert(ids2))


def synthetic_token_batch(B=8, T=16, vocab=64):
    """Copy-task trajectories with terminal reward for matching token 0."""
    ids = torch.randint(1, vocab, (B, T))
    ids[:, T // 2 :] = ids[:, :1]
    rewards = torch.zeros(B, T - 1)
    rewards[:, -1] = 1.0
    mask = torch.ones(B, T - 1)
    return {"ids": ids, "rewards": rewards, "mask": mask}


class TestILQL:
    def test_learn_and_generate(self):
        torch.manual_seed(0)
        agent = ILQL(vocab_size=64, n_layer=2, n_head=4, n_embd=32, max_positions=32, lr=1e-3)
        first = agent.learn(synthetic_token_batch())
        for _ in range(5):
            stats = agent.learn(synthetic_token_batch())
        assert np.isfinite(stats["loss"])
        assert np.isfinite(stats["cql"])
        out = agent.generate(torch.randint(1, 64, (2, 4)), max_new_tokens=4, beta=1.0)
        assert out.shape == (2, 8)

    def test_target_heads_track(self):
        agent = ILQL(vocab_size=64, n_layer=1, n_head=2, n_embd=16, max_positions=32, polyak=1.0)
        agent.learn(synthetic_token_batch(vocab=64))
        torch.testing.assert_close(
            agent.q1_target.weight, agent.q1_head.weight, rtol=1e-5, atol=1e-6
        )


class TestBCLM:
    def test_bc_reduces_loss(self):
        torch.manual_seed(0)
        agent = BC_LM(vocab_size=64, n_layer=2, n_head=4, n_embd=32, max_positions=32, lr=3e-3)
        batch = synthetic_token_batch()
        first = agent.learn(batch)["loss"]
        for _ in range(20):
            last = agent.learn(batch)["loss"]
        assert last < first

    def test_top_p_sampling(self):
        agent = BC_LM(vocab_size=64, n_layer=1, n_head=2, n_embd=16, max_positions=32)
        out = agent.generate(torch.randint(1, 64, (2, 4)), 4, top_p=0.9, top_k=10)
        assert out.shape == (2, 8)
