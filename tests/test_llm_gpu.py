"""GPU LLM-path tests (tiny random Llama, bf16, fused kernels)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"

TINY = dict(
    model_type="llama", vocab_size=512, hidden_size=128, intermediate_size=256,
    num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
    max_position_embeddings=512, pad_token_id=0,
)


class TestFusedLogprobsGpu:
    def test_bf16_matches_fp32_reference(self):
        from agilerl_amd.ops.fused_logprobs import fused_linear_logprobs

        N, H, V = 64, 128, 32000
        hidden = torch.randn(N, H, device=DEV, dtype=torch.bfloat16, requires_grad=True)
        weight = torch.randn(V, H, device=DEV, dtype=torch.bfloat16, requires_grad=True)
        targets = torch.randint(0, V, (N,), device=DEV)
        lp = fused_linear_logprobs(hidden, weight, targets, chunk_rows=32)
        ref = torch.log_softmax((hidden.float() @ weight.float().t()), -1).gather(
            1, targets.unsqueeze(1)
        ).squeeze(1)
        torch.testing.assert_close(lp, ref, rtol=5e-2, atol=5e-2)

        g = torch.randn(N, device=DEV)
        gh, gw = torch.autograd.grad(lp, [hidden, weight], g, retain_graph=True)
        h2 = hidden.detach().float().requires_grad_(True)
        w2 = weight.detach().float().requires_grad_(True)
        ref2 = torch.log_softmax(h2 @ w2.t(), -1).gather(1, targets.unsqueeze(1)).squeeze(1)
        rgh, rgw = torch.autograd.grad(ref2, [h2, w2], g)
        torch.testing.assert_close(gh.float(), rgh, rtol=0.1, atol=0.1)


class TestGRPOGpu:
    def test_grpo_learn_bf16(self):
        from agilerl_amd.algorithms.llm.grpo import GRPO
        from agilerl_amd.llm_envs import TokenReasoningGym, make_grpo_experiences

        agent = GRPO(
            model_config=dict(TINY), dtype=torch.bfloat16,
            lora_config={"r": 4}, group_size=4, micro_batch_size=8,
            beta=0.04, lr=1e-4, max_completion_tokens=16, device=DEV,
        )
        env = TokenReasoningGym(vocab_size=512, prompt_len=16, data_batch_size=2, group_size=4)
        prompts = env.reset()
        seqs = agent.get_action(prompts)
        rewards = env.score(seqs)
        exp = make_grpo_experiences(env, seqs, rewards)
        stats = agent.learn(exp)
        assert np.isfinite(stats["loss"])
        # adapter params actually moved
        norms = [float(p.float().norm()) for p in agent.policy_network.parameters()]
        assert any(n > 0 for n in norms)

    def test_sft_gpu(self):
        from agilerl_amd.algorithms.llm.sft import SFT
        from agilerl_amd.llm_envs import SyntheticSFTGym

        agent = SFT(model_config=dict(TINY), dtype=torch.bfloat16,
                    lora_config={"r": 4}, micro_batch_size=8, lr=1e-3, device=DEV)
        env = SyntheticSFTGym(vocab_size=512, prompt_len=8, completion_len=8,
                              data_batch_size=16, seed=0)
        first = agent.learn(env.sample())["loss"]
        for _ in range(10):
            last = agent.learn(env.sample())["loss"]
        assert last < first


class TestHipRMSNorm:
    def test_matches_eager(self):
        from agilerl_amd.ops.rmsnorm import rms_norm

        x = torch.randn(37, 4096, device=DEV, dtype=torch.bfloat16, requires_grad=True)
        w = torch.randn(4096, device=DEV, dtype=torch.bfloat16, requires_grad=True)
        y = rms_norm(x, w, 1e-6)
        xf = x.float()
        ref = (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()).to(torch.bfloat16)
        torch.testing.assert_close(y, ref, rtol=2e-2, atol=2e-2)

        g = torch.randn_like(y)
        gx, gw = torch.autograd.grad(y, [x, w], g, retain_graph=True)
        x2 = x.detach().float().requires_grad_(True)
        w2 = w.detach().float().requires_grad_(True)
        ref2 = x2 * torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-6) * w2
        rgx, rgw = torch.autograd.grad(ref2, [x2, w2], g.float())
        torch.testing.assert_close(gx.float(), rgx, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(gw.float(), rgw, rtol=5e-2, atol=2e-1)

    def test_llama_patch(self):
        from agilerl_amd.architectures import apply_hip_kernels_to_llama
        from transformers import AutoConfig, AutoModelForCausalLM

        cfg = AutoConfig.for_model("llama", vocab_size=256, hidden_size=128,
                                   intermediate_size=256, num_hidden_layers=2,
                                   num_attention_heads=4, num_key_value_heads=2,
                                   max_position_embeddings=128, pad_token_id=0)
        m = AutoModelForCausalLM.from_config(cfg, dtype=torch.bfloat16).to(DEV)
        ids = torch.randint(0, 256, (2, 16), device=DEV)
        with torch.no_grad():
            ref = m(input_ids=ids).logits.float()
        n = apply_hip_kernels_to_llama(m)
        assert n >= 5  # per-layer input/post-attn + final norm
        with torch.no_grad():
            out = m(input_ids=ids).logits.float()
        torch.testing.assert_close(out, ref, rtol=5e-2, atol=5e-1)


class TestPackingEquivalenceGpu:
    def test_packed_matches_padded_bf16(self):
        """compute_logprobs_packed == compute_logprobs on real positions for
        a ragged bf16 batch on GPU (packing is default-'auto' now)."""
        from agilerl_amd.algorithms.llm.grpo import GRPO

        tiny = dict(model_type="llama", vocab_size=128, hidden_size=64,
                    intermediate_size=128, num_hidden_layers=2,
                    num_attention_heads=4, num_key_value_heads=2,
                    max_position_embeddings=256, pad_token_id=0)
        torch.manual_seed(0)
        agent = GRPO(model_config=tiny, dtype=torch.bfloat16,
                     lora_config={"r": 4}, device="cuda:0")
        B, T = 6, 24
        g = torch.Generator().manual_seed(1)
        ids = torch.randint(1, 128, (B, T), generator=g).to("cuda:0")
        mask = torch.ones_like(ids)
        for i, pad in enumerate([0, 3, 7, 11, 5, 16]):  # ragged left pads
            ids[i, :pad] = 0
            mask[i, :pad] = 0
        lp_pad = agent.compute_logprobs(ids, mask)
        lp_pack = agent.compute_logprobs_packed(ids, mask)
        real = mask[:, 1:].bool() & mask[:, :-1].bool()
        torch.testing.assert_close(
            lp_pack[real].float(), lp_pad[real].float(), rtol=5e-2, atol=5e-1)

    def test_auto_packing_selects_by_pad_fraction(self):
        from agilerl_amd.algorithms.llm.grpo import GRPO

        tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                    intermediate_size=64, num_hidden_layers=1,
                    num_attention_heads=2, num_key_value_heads=1,
                    max_position_embeddings=128, pad_token_id=0)
        agent = GRPO(model_config=tiny, dtype=torch.bfloat16,
                     lora_config={"r": 2}, device="cuda:0", group_size=2)
        assert agent.use_packing == "auto"
        # heavily padded batch learns fine through the packed path
        ids = torch.randint(1, 64, (4, 24)).to("cuda:0")
        mask = torch.ones_like(ids)
        mask[:, :14] = 0
        ids[:, :14] = 0
        pos = torch.arange(23, device="cuda:0").unsqueeze(0)
        action_mask = ((pos + 1 >= 18) & (mask[:, 1:].bool())).float().expand(4, 23)
        stats = agent.learn({"ids": ids, "attention_mask": mask,
                             "action_mask": action_mask,
                             "rewards": torch.rand(4)})
        assert np.isfinite(stats["loss"])
