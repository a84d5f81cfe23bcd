"""Paged KV cache + continuous-batching decode engine (eager parity)."""

import numpy as np
import pytest
import torch

from agilerl_amd.llm.decode_engine import DecodeEngine
from agilerl_amd.llm.paged_cache import PagedKVCache


def tiny_model():
    from transformers import AutoConfig, AutoModelForCausalLM

    torch.manual_seed(0)
    cfg = AutoConfig.for_model(
        "llama", vocab_size=64, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=128, pad_token_id=0,
    )
    return AutoModelForCausalLM.from_config(cfg).eval()


class TestPagedKVCache:
    def test_append_gather_roundtrip(self):
        cache = PagedKVCache(num_layers=2, num_kv_heads=1, head_dim=4,
                             num_pages=8, page_size=4)
        cache.alloc(0)
        k = torch.arange(2 * 6 * 1 * 4, dtype=torch.float32).reshape(2, 6, 1, 4)
        cache.append(0, k, k * 2)  # 6 tokens spans 2 pages
        gk, gv, mask = cache.gather([0])
        assert gk.shape == (2, 1, 1, 6, 4)
        torch.testing.assert_close(gk[:, 0].permute(0, 2, 1, 3), k)
        torch.testing.assert_close(gv[:, 0].permute(0, 2, 1, 3), k * 2)
        assert mask.sum() == 6

    def test_left_padding_and_free(self):
        cache = PagedKVCache(2, 1, 4, num_pages=8, page_size=4)
        cache.alloc(0), cache.alloc(1)
        cache.append(0, torch.ones(2, 2, 1, 4), torch.ones(2, 2, 1, 4))
        cache.append(1, torch.ones(2, 7, 1, 4), torch.ones(2, 7, 1, 4))
        gk, _, mask = cache.gather([0, 1])
        assert gk.shape[3] == 7
        assert (mask[0] == torch.tensor([0, 0, 0, 0, 0, 1, 1])).all()
        free_before = cache.free_pages
        cache.free(1)
        assert cache.free_pages == free_before + 2
        with pytest.raises(RuntimeError):
            big = torch.ones(2, 100, 1, 4)
            cache.append(0, big, big)


class TestDecodeEngine:
    def _reference(self, model, prompt, n):
        out = model.generate(
            input_ids=prompt.unsqueeze(0), max_new_tokens=n, do_sample=False,
            pad_token_id=0, use_cache=True,
        )
        return out[0]

    def test_greedy_parity_ragged_batch(self):
        model = tiny_model()
        engine = DecodeEngine(model, num_pages=64, page_size=4)
        torch.manual_seed(1)
        prompts = [torch.randint(1, 64, (n,)) for n in (3, 7, 5)]
        ids = [engine.submit(p, max_new_tokens=6) for p in prompts]
        results = engine.run_all()
        for sid, prompt in zip(ids, prompts):
            expected = self._reference(model, prompt, 6)
            torch.testing.assert_close(results[sid], expected)

    def test_continuous_admission_mid_flight(self):
        model = tiny_model()
        engine = DecodeEngine(model, num_pages=64, page_size=4)
        torch.manual_seed(2)
        p1 = torch.randint(1, 64, (4,))
        p2 = torch.randint(1, 64, (6,))
        p3 = torch.randint(1, 64, (5,))
        s1 = engine.submit(p1, max_new_tokens=8)
        s2 = engine.submit(p2, max_new_tokens=4)
        results = {}
        for _ in range(3):
            results.update(dict(engine.step()))
        s3 = engine.submit(p3, max_new_tokens=5)  # admitted mid-flight
        results.update(engine.run_all())
        for sid, prompt, n in ((s1, p1, 8), (s2, p2, 4), (s3, p3, 5)):
            torch.testing.assert_close(results[sid], self._reference(model, prompt, n))

    def test_pages_freed_and_queueing(self):
        model = tiny_model()
        # tiny pool: only one sequence fits at a time
        engine = DecodeEngine(model, num_pages=3, page_size=4)
        torch.manual_seed(3)
        prompts = [torch.randint(1, 64, (4,)) for _ in range(3)]
        ids = [engine.submit(p, max_new_tokens=4) for p in prompts]
        results = engine.run_all()
        assert set(results) == set(ids)
        assert engine.cache.free_pages == 3
        for sid, p in zip(ids, prompts):
            torch.testing.assert_close(results[sid], self._reference(model, p, 4))

    def test_temperature_sampling_runs(self):
        model = tiny_model()
        engine = DecodeEngine(model, num_pages=32, page_size=4)
        engine.submit(torch.randint(1, 64, (4,)), max_new_tokens=4, temperature=1.0)
        results = engine.run_all()
        assert len(results) == 1


class TestGRPOPagedGeneration:
    def test_generate_paged_matches_generate_greedy(self):
        from agilerl_amd.algorithms.llm.grpo import GRPO

        tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                    intermediate_size=64, num_hidden_layers=1,
                    num_attention_heads=2, num_key_value_heads=1,
                    max_position_embeddings=128, pad_token_id=0)
        torch.manual_seed(0)
        agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                     max_completion_tokens=6)
        ids = torch.randint(1, 64, (3, 5))
        ids[0, :2] = 0  # left-padded row
        mask = (ids != 0).long()
        ref = agent.generate(ids, mask, do_sample=False)
        paged = agent.generate_paged(ids, mask, do_sample=False)
        torch.testing.assert_close(paged, ref)

    def test_generate_paged_sampling_shape(self):
        from agilerl_amd.algorithms.llm.grpo import GRPO

        tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                    intermediate_size=64, num_hidden_layers=1,
                    num_attention_heads=2, num_key_value_heads=1,
                    max_position_embeddings=128, pad_token_id=0)
        agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                     max_completion_tokens=4)
        ids = torch.randint(1, 64, (4, 5))
        mask = torch.ones_like(ids)
        out = agent.generate_paged(ids, mask, do_sample=True)
        assert out.shape == (4, 9)
        # engine reused on second call
        eng = agent._decode_engine
        agent.generate_paged(ids, mask, do_sample=True)
        assert agent._decode_engine is eng


def test_grpo_full_loop_with_paged_generation():
    import numpy as np

    from agilerl_amd.algorithms.llm.grpo import GRPO
    from agilerl_amd.llm_envs import TokenReasoningGym, make_grpo_experiences

    tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                intermediate_size=64, num_hidden_layers=1,
                num_attention_heads=2, num_key_value_heads=1,
                max_position_embeddings=128, pad_token_id=0)
    torch.manual_seed(0)
    agent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                 group_size=2, max_completion_tokens=4, generation="paged")
    env = TokenReasoningGym(vocab_size=64, prompt_len=4, data_batch_size=2,
                            group_size=2, seed=0)
    prompts = env.reset()
    seqs = agent.get_action(prompts)
    assert seqs.shape == (4, 8)
    stats = agent.learn(make_grpo_experiences(env, seqs, env.score(seqs)))
    assert np.isfinite(stats["loss"])


def test_paged_attention_eager_reference_cpu():
    """Eager paged-attention equals dense SDPA on gathered K/V."""
    import math

    from agilerl_amd.ops.paged_attn import paged_attention_decode

    torch.manual_seed(0)
    B, Hq, Hkv, D, S, P = 2, 4, 2, 16, 4, 8
    q = torch.randn(B, Hq, D)
    kp = torch.randn(P, S, Hkv, D)
    vp = torch.randn(P, S, Hkv, D)
    table = torch.tensor([[0, 2, 4, 6], [1, 3, 5, 7]], dtype=torch.int32)
    lengths = torch.tensor([13, 9])
    out = paged_attention_decode(q, kp, vp, table, lengths)
    for b in range(B):
        n = int(lengths[b])
        pages = table[b, : (n + S - 1) // S].long()
        k = kp[pages].reshape(-1, Hkv, D)[:n]
        v = vp[pages].reshape(-1, Hkv, D)[:n]
        gqa = Hq // Hkv
        for h in range(Hq):
            hk = h // gqa
            att = torch.softmax((k[:, hk] @ q[b, h]) / math.sqrt(D), 0)
            torch.testing.assert_close(out[b, h], att @ v[:, hk], atol=1e-5, rtol=1e-5)


def test_engine_multi_adapter_population_sharing():
    """One engine serves a whole GRPO population: per-sequence adapters
    are grouped per step and activated through set_adapter_fn."""
    from agilerl_amd.algorithms.llm.grpo import GRPO

    tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                intermediate_size=64, num_hidden_layers=1,
                num_attention_heads=2, num_key_value_heads=1,
                max_position_embeddings=128, pad_token_id=0)
    torch.manual_seed(0)
    pop = GRPO.population(2, model_config=tiny, dtype=torch.float32,
                          lora_config={"r": 2}, max_completion_tokens=4)
    a0, a1 = pop
    assert a0.model is a1.model
    activations = []

    def set_adapter(name):
        activations.append(name)
        from agilerl_amd.llm import set_active_adapter

        set_active_adapter(a0.model, name)

    engine = DecodeEngine(a0.model, num_pages=64, page_size=4,
                          set_adapter_fn=set_adapter)
    p = torch.randint(1, 64, (4,))
    s0 = engine.submit(p, max_new_tokens=3, adapter=a0.adapter_name)
    s1 = engine.submit(p, max_new_tokens=3, adapter=a1.adapter_name)
    results = engine.run_all()
    assert set(results) == {s0, s1}
    assert a0.adapter_name in activations and a1.adapter_name in activations


def test_engine_stats():
    model = tiny_model()
    engine = DecodeEngine(model, num_pages=32, page_size=4)
    engine.submit(torch.randint(1, 64, (4,)), max_new_tokens=3)
    engine.run_all()
    st = engine.stats()
    assert st["sequences_finished"] == 1
    assert st["tokens_generated"] == 3
    assert st["free_pages"] == 32
    assert st["tokens_per_sec"] >= 0


def test_engine_cancel():
    model = tiny_model()
    engine = DecodeEngine(model, num_pages=32, page_size=4)
    s1 = engine.submit(torch.randint(1, 64, (4,)), max_new_tokens=8)
    s2 = engine.submit(torch.randint(1, 64, (4,)), max_new_tokens=8)
    engine.step()  # both active
    assert engine.cancel(s1)
    results = engine.run_all()
    assert s1 not in results and s2 in results
    assert engine.cache.free_pages == 32
    assert not engine.cancel(999)


def test_eos_stops_sequence_early():
    model = tiny_model()
    torch.manual_seed(4)
    prompt = torch.randint(1, 64, (5,))
    # discover the greedy first token, then make it the EOS
    probe = DecodeEngine(model, num_pages=32, page_size=4)
    sid = probe.submit(prompt, max_new_tokens=1)
    first = int(probe.run_all()[sid][-1])

    engine = DecodeEngine(model, num_pages=32, page_size=4, eos_token_id=first)
    sid = engine.submit(prompt, max_new_tokens=8)
    out = engine.run_all()[sid]
    assert out.numel() == prompt.numel() + 1  # stopped at EOS, not max tokens
    assert int(out[-1]) == first


class TestCloneDecodeEngineIsolation:
    def test_clone_does_not_share_parent_engine(self):
        """copy.copy shared _decode_engine whose set_adapter_fn was
        bound to the PARENT's _activate — a clone generating paged would
        sample under the parent's adapter.  The clone must drop the engine
        and lazily rebuild one bound to its own _activate."""
        from agilerl_amd.algorithms.llm.grpo import GRPO

        tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                    intermediate_size=64, num_hidden_layers=1,
                    num_attention_heads=2, num_key_value_heads=1,
                    max_position_embeddings=128, pad_token_id=0)
        torch.manual_seed(0)
        parent = GRPO(model_config=tiny, dtype=torch.float32, lora_config={"r": 2},
                      max_completion_tokens=4)
        ids = torch.randint(1, 64, (2, 5))
        mask = torch.ones_like(ids)
        parent.generate_paged(ids, mask, do_sample=False)
        assert parent._decode_engine is not None
        clone = parent.clone(index=7)
        assert getattr(clone, "_decode_engine", None) is None
        # clone's lazily-built engine must activate the CLONE's adapter
        clone.generate_paged(ids, mask, do_sample=False)
        assert clone._decode_engine is not parent._decode_engine
        assert clone._decode_engine.set_adapter_fn.__self__ is clone


class TestPagedAttentionDecodePath:
    """The paged-attention decode path (per-layer step + pool-direct
    attention) must match model.generate greedy exactly — run on CPU via
    the eager reference attention, so the runner math (RoPE, norms,
    residuals, pool writes) is pinned without a GPU."""

    def _model(self, seed=0):
        import torch as _t
        from transformers import AutoConfig, AutoModelForCausalLM

        cfg = AutoConfig.for_model(
            "llama", vocab_size=64, hidden_size=32, intermediate_size=64,
            num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
            max_position_embeddings=128, pad_token_id=0)
        _t.manual_seed(seed)
        return AutoModelForCausalLM.from_config(cfg)

    def test_greedy_parity_vs_generate(self):
        model = self._model()
        engine = DecodeEngine(model, num_pages=64, page_size=4,
                              use_paged_attention=True)
        assert engine._paged_runner is not None
        torch.manual_seed(1)
        prompts = [torch.randint(1, 64, (n,)) for n in (7, 3, 5)]
        sids = [engine.submit(p, max_new_tokens=8) for p in prompts]
        results = engine.run_all()
        for p, sid in zip(prompts, sids):
            ref = model.generate(
                p.unsqueeze(0), max_new_tokens=8, do_sample=False,
                pad_token_id=0)
            torch.testing.assert_close(results[sid], ref[0])

    def test_ragged_admission_parity(self):
        model = self._model(seed=3)
        engine = DecodeEngine(model, max_batch=2, num_pages=64, page_size=4,
                              use_paged_attention=True)
        torch.manual_seed(2)
        prompts = [torch.randint(1, 64, (n,)) for n in (6, 4, 9, 2)]
        sids = [engine.submit(p, max_new_tokens=5) for p in prompts]
        results = engine.run_all()
        for p, sid in zip(prompts, sids):
            ref = model.generate(p.unsqueeze(0), max_new_tokens=5,
                                 do_sample=False, pad_token_id=0)
            torch.testing.assert_close(results[sid], ref[0])
