"""EvolvableGPT / EvolvableBERT / ILQL / BC_LM tests."""

import numpy as np
import pytest
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


class TestSamplingUtils:
    def test_top_k_masks_tail(self):
        from agilerl_amd.utils.sampling import process_logits

        logits = torch.tensor([[1.0, 5.0, 3.0, 0.5, 4.0]])
        out = process_logits(logits, top_k=2)
        keep = torch.isfinite(out[0])
        assert keep.tolist() == [False, True, False, False, True]

    def test_top_p_keeps_nucleus(self):
        from agilerl_amd.utils.sampling import process_logits

        logits = torch.log(torch.tensor([[0.5, 0.3, 0.15, 0.05]]))
        out = process_logits(logits, top_p=0.7)
        keep = torch.isfinite(out[0])
        # 0.5 alone < 0.7, so 0.3 is kept too; 0.15/0.05 dropped
        assert keep.tolist() == [True, True, False, False]
        # softmax renormalizes over the kept set
        probs = torch.softmax(out, dim=-1)[0]
        assert probs[0].item() == pytest.approx(0.5 / 0.8, rel=1e-5)

    def test_sample_respects_mask(self):
        from agilerl_amd.utils.sampling import sample_from_logits

        torch.manual_seed(0)
        logits = torch.tensor([[10.0, -1.0, -1.0, -1.0]]).expand(64, 4)
        out = sample_from_logits(logits, top_k=1)
        assert (out == 0).all()


class TestILQLPolicyEvaluator:
    def _model(self):
        torch.manual_seed(0)
        from agilerl_amd.algorithms.ilql import ILQL

        return ILQL(vocab_size=32, n_layer=1, n_head=2, n_embd=32,
                    max_positions=64)

    def test_policy_act_and_eos_stop(self):
        from agilerl_amd.algorithms.ilql import ILQL_Policy

        model = self._model()
        pol = ILQL_Policy(model, beta=0.5, max_new_tokens=6, eos_token_id=3,
                          greedy=True)
        ctx = torch.randint(4, 32, (2, 5))
        comp = pol.act(ctx)
        assert comp.shape[0] == 2 and 1 <= comp.shape[1] <= 6
        # after an EOS, subsequent tokens are EOS (frozen rows)
        for row in comp:
            hit = (row == 3).nonzero()
            if hit.numel():
                assert (row[int(hit[0]):] == 3).all()

    def test_evaluator_scores_token_env(self):
        from agilerl_amd.algorithms.ilql import ILQL_Evaluator, ILQL_Policy

        class CountTargetEnv:
            def reset(self):
                return torch.randint(4, 32, (3, 4))

            def score(self, full):
                return (full[:, 4:] == 7).float().mean(dim=1).numpy()

        model = self._model()
        pol = ILQL_Policy(model, max_new_tokens=5)
        ev = ILQL_Evaluator(CountTargetEnv(), n_batches=2)
        stats = ev.evaluate(pol)
        assert set(stats) == {"mean_reward", "std_reward", "mean_value", "n"}
        assert stats["n"] == 6
        assert 0.0 <= stats["mean_reward"] <= 1.0

    def test_beta_changes_decode_distribution(self):
        """beta=0 ignores Q-V; large beta shifts decoding toward high-Q
        tokens — distributions must differ."""
        model = self._model()
        torch.manual_seed(1)
        ctx = torch.randint(4, 32, (8, 6))
        torch.manual_seed(2)
        out0 = model.generate(ctx, 8, beta=0.0)
        torch.manual_seed(2)
        outb = model.generate(ctx, 8, beta=50.0)
        assert not torch.equal(out0, outb)


class TestSeq2SeqBERT:
    """Reference modules/bert.py:63 end-to-end encoder-decoder surface."""

    def _s2s(self):
        return EvolvableBERT(encoder_layers=[64, 64], decoder_layers=[64],
                             src_vocab_size=50, tgt_vocab_size=40, d_model=32,
                             n_head=4, max_positions=32)

    def test_forward_and_loss(self):
        m = self._s2s()
        src = torch.randint(0, 50, (2, 10))
        tgt = torch.randint(0, 40, (2, 7))
        assert m(src, tgt).shape == (2, 7, 40)
        _, loss = m(src, tgt, targets=tgt)
        loss.backward()
        assert torch.isfinite(loss)

    def test_decoder_causal_encoder_bidirectional(self):
        m = self._s2s()
        m.eval()
        src = torch.randint(0, 50, (1, 10))
        tgt = torch.randint(0, 40, (1, 6))
        t2 = tgt.clone()
        t2[0, -1] = (t2[0, -1] + 1) % 40
        a, b = m(src, tgt), m(src, t2)
        assert torch.allclose(a[0, :4], b[0, :4], atol=1e-5)   # causal decoder
        s2 = src.clone()
        s2[0, -1] = (s2[0, -1] + 1) % 50
        c = m(s2, tgt)
        assert not torch.allclose(a[0, 0], c[0, 0])  # encoder sees all of src

    def test_layer_mutations_preserve_function_shape(self):
        m = self._s2s()
        src = torch.randint(0, 50, (2, 8))
        tgt = torch.randint(0, 40, (2, 5))
        for mut in ("add_encoder_layer", "add_decoder_layer", "add_node",
                    "remove_encoder_layer"):
            m.apply_mutation(mut)
            assert m(src, tgt).shape == (2, 5, 40)
        c = m.clone()
        m.eval(), c.eval()
        torch.testing.assert_close(c(src, tgt), m(src, tgt))


class TestTokenRLDataStack:
    """Reference agilerl/data/* stack: dialogue -> DataPoint -> ILQL batch
    (rl_data.py:53), interaction loop (language_environment.py:56)."""

    class Dialogue:
        def __init__(self, turns, terminal=True):
            self.turns, self.terminal = turns, terminal

        def to_sequence(self):
            return self.turns, self.terminal

        def metadata(self):
            return None

    def _tok(self):
        from agilerl_amd.data import DialogueTokenizer

        return DialogueTokenizer(vocab_size=123)

    def test_datapoint_extraction(self):
        from agilerl_amd.data import DataPoint

        tok = self._tok()
        obs = self.Dialogue([("hi", None), ("yes", 1.5), ("ok", None), ("no", -0.5)])
        dp = DataPoint.from_obs(obs, tok)
        assert abs(sum(dp.rewards) - 1.0) < 1e-6          # 1.5 - 0.5
        assert dp.utterance_rewards == [1.5, -0.5]
        assert dp.terminals[-1] == 1
        # round-trip through the tokenizer
        assert tok.decode(dp.tokens) == dp.raw_str

    def test_token_reward_shaping(self):
        from agilerl_amd.data import DataPoint, SpecifiedTokenReward

        tok = self._tok()
        obs = self.Dialogue([("a", None), ("bb", 2.0)])
        shaped = SpecifiedTokenReward({ord("b"): 0.1})
        dp = DataPoint.from_obs(obs, tok, shaped)
        # action span covers <eos>+2 b-tokens; the two b rewards + final 2.0
        assert abs(sum(dp.rewards) - 2.2) < 1e-6

    def test_dataset_feeds_ilql(self):
        import numpy as np

        from agilerl_amd.algorithms.ilql import ILQL
        from agilerl_amd.data import ListRLDataset

        tok = self._tok()
        obs = [self.Dialogue([("q", None), ("a", 1.0)]) for _ in range(4)]
        ds = ListRLDataset(obs, tok, max_len=16)
        batch = ds.sample_batch(4)
        assert batch["ids"].shape == (4, 16)
        assert batch["mask"].sum() > 0
        agent = ILQL(vocab_size=tok.vocab_size, n_embd=32, n_head=2,
                     n_layer=1, max_positions=32)
        stats = agent.learn(batch)
        assert np.isfinite(stats["loss"])

    def test_interact_environment(self):
        from agilerl_amd.data import (LanguageEnvironment, LanguagePolicy,
                                      interact_environment)

        outer = self

        class EchoEnv(LanguageEnvironment):
            def __init__(self):
                self.n = 0

            def reset(self):
                self.n = 0
                return outer.Dialogue([("q", None)], terminal=False)

            def step(self, action):
                self.n += 1
                done = self.n >= 3
                return outer.Dialogue([("q", None), (action, 1.0)], terminal=done), 1.0, done

        class Fixed(LanguagePolicy):
            def act(self, obs):
                return "a"

        _, total, turns = interact_environment(EchoEnv(), Fixed())
        assert total == 3.0 and turns == 3
