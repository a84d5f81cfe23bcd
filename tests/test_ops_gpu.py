"""HIP kernel numerics vs eager fp32 PyTorch references (gpu-marked)."""

import numpy as np
import pytest
import torch

from agilerl_amd import ops
from agilerl_amd.ops.backend import extension

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture(scope="module", autouse=True)
def _require_ext():
    assert extension() is not None, "HIP extension must be built on GPU boxes"


class TestGaeScan:
    def test_matches_cpu(self):
        T, N = 64, 129
        rewards = torch.randn(T, N)
        values = torch.randn(T, N)
        dones = (torch.rand(T, N) < 0.1).float()
        last_value = torch.randn(N)
        adv_cpu, ret_cpu = ops.gae_scan(rewards, values, dones, last_value, 0.99, 0.95)
        adv_gpu, ret_gpu = ops.gae_scan(
            rewards.to(DEV), values.to(DEV), dones.to(DEV), last_value.to(DEV), 0.99, 0.95
        )
        torch.testing.assert_close(adv_gpu.cpu(), adv_cpu, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(ret_gpu.cpu(), ret_cpu, rtol=1e-4, atol=1e-5)


class TestNStep:
    def test_matches_cpu(self):
        B, n = 257, 5
        rewards = torch.randn(B, n)
        dones = (torch.rand(B, n) < 0.2).float()
        r_cpu, s_cpu = ops.nstep_scan(rewards, dones, 0.97)
        r_gpu, s_gpu = ops.nstep_scan(rewards.to(DEV), dones.to(DEV), 0.97)
        torch.testing.assert_close(r_gpu.cpu(), r_cpu, rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(s_gpu.cpu(), s_cpu)


class TestC51:
    def test_matches_cpu(self):
        B, A = 128, 51
        dist = torch.softmax(torch.randn(B, A), -1)
        rewards = torch.randn(B) * 5
        dones = (torch.rand(B) < 0.3).float()
        support = torch.linspace(-10, 10, A)
        p_cpu = ops.c51_project(dist, rewards, dones, support, 0.99, -10, 10)
        p_gpu = ops.c51_project(
            dist.to(DEV), rewards.to(DEV), dones.to(DEV), support.to(DEV), 0.99, -10, 10
        )
        torch.testing.assert_close(p_gpu.cpu(), p_cpu, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(p_gpu.sum(-1).cpu(), torch.ones(B), rtol=1e-4, atol=1e-5)


class TestPolyak:
    def test_matches_foreach(self):
        tgt = [torch.randn(100, 37, device=DEV), torch.randn(5, device=DEV), torch.randn(64, device=DEV)]
        src = [torch.randn_like(t) for t in tgt]
        expected = [t + 0.1 * (s - t) for t, s in zip(tgt, src)]
        ops.polyak_update_(tgt, src, 0.1)
        for t, e in zip(tgt, expected):
            torch.testing.assert_close(t, e, rtol=1e-5, atol=1e-6)


class TestNoisyLinear:
    def test_forward_backward_match(self):
        torch.manual_seed(0)
        B, I, O = 64, 37, 29
        x = torch.randn(B, I, device=DEV, requires_grad=True)
        w_mu = torch.randn(O, I, device=DEV, requires_grad=True)
        w_sig = torch.rand(O, I, device=DEV, requires_grad=True)
        w_eps = torch.randn(O, I, device=DEV)
        b_mu = torch.randn(O, device=DEV, requires_grad=True)
        b_sig = torch.rand(O, device=DEV, requires_grad=True)
        b_eps = torch.randn(O, device=DEV)

        out = ops.noisy_linear(x, w_mu, w_sig, w_eps, b_mu, b_sig, b_eps)
        ref = torch.nn.functional.linear(x, w_mu + w_sig * w_eps, b_mu + b_sig * b_eps)
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)

        g = torch.randn_like(out)
        grads = torch.autograd.grad(out, [x, w_mu, w_sig, b_mu, b_sig], g, retain_graph=True)
        ref_grads = torch.autograd.grad(ref, [x, w_mu, w_sig, b_mu, b_sig], g)
        for a, b in zip(grads, ref_grads):
            torch.testing.assert_close(a, b, rtol=1e-3, atol=1e-4)


class TestGroupAdvantage:
    def test_matches_cpu(self):
        r = torch.randn(16 * 8)
        a_cpu = ops.group_advantage(r, 8)
        a_gpu = ops.group_advantage(r.to(DEV), 8)
        torch.testing.assert_close(a_gpu.cpu(), a_cpu, rtol=1e-4, atol=1e-5)

    def test_large_group(self):
        r = torch.randn(4 * 300)
        a_cpu = ops.group_advantage(r, 300)
        a_gpu = ops.group_advantage(r.to(DEV), 300)
        torch.testing.assert_close(a_gpu.cpu(), a_cpu, rtol=1e-4, atol=1e-4)


class TestRowLseGather:
    def test_matches_eager(self):
        ext = extension()
        rows, V = 128, 32000
        logits = torch.randn(rows, V, device=DEV)
        targets = torch.randint(0, V, (rows,), device=DEV)
        lp, lse = ext.row_lse_gather(logits, targets, 1.0)
        ref_lse = torch.logsumexp(logits, dim=-1)
        ref_lp = logits.gather(1, targets.unsqueeze(1)).squeeze(1) - ref_lse
        torch.testing.assert_close(lse, ref_lse, rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(lp, ref_lp, rtol=1e-4, atol=1e-4)

    def test_temperature(self):
        ext = extension()
        rows, V = 16, 1000
        logits = torch.randn(rows, V, device=DEV)
        targets = torch.randint(0, V, (rows,), device=DEV)
        lp, _ = ext.row_lse_gather(logits, targets, 2.0)
        scaled = logits / 2.0
        ref = scaled.gather(1, targets.unsqueeze(1)).squeeze(1) - torch.logsumexp(scaled, -1)
        torch.testing.assert_close(lp, ref, rtol=1e-4, atol=1e-4)

    def test_softmax_bwd(self):
        ext = extension()
        rows, V = 32, 5000
        logits = torch.randn(rows, V, device=DEV, requires_grad=True)
        targets = torch.randint(0, V, (rows,), device=DEV)
        g = torch.randn(rows, device=DEV)
        lse_ref = torch.logsumexp(logits, -1)
        lp_ref = logits.gather(1, targets.unsqueeze(1)).squeeze(1) - lse_ref
        (ref_grad,) = torch.autograd.grad(lp_ref, logits, g)

        work = logits.detach().clone()
        ext.row_softmax_bwd_(work, targets, lse_ref.detach(), g, 1.0)
        torch.testing.assert_close(work, ref_grad, rtol=1e-3, atol=1e-5)


class TestGrpoTokenLoss:
    def _eager(self, logp, old_logp, ref_logp, adv, mask, lo, hi, kl, cispo):
        logp = logp.detach().requires_grad_(True)
        ratio = (logp - old_logp).exp()
        if cispo:
            w = ratio.clamp(lo, hi).detach()
            loss = -w * adv * logp
        else:
            loss = -torch.minimum(ratio * adv, ratio.clamp(lo, hi) * adv)
        if ref_logp is not None and kl:
            d = ref_logp - logp
            loss = loss + kl * (d.exp() - d - 1)
        loss = loss * mask
        (dl,) = torch.autograd.grad(loss.sum(), logp)
        return loss.detach(), dl

    @pytest.mark.parametrize("cispo", [False, True])
    @pytest.mark.parametrize("use_ref", [False, True])
    def test_matches_eager(self, cispo, use_ref):
        ext = extension()
        N = 4096
        logp = torch.randn(N, device=DEV) * 0.5
        old = logp + torch.randn(N, device=DEV) * 0.2
        ref = logp + torch.randn(N, device=DEV) * 0.2 if use_ref else None
        adv = torch.randn(N, device=DEV)
        mask = (torch.rand(N, device=DEV) < 0.8).float()
        loss, dl = ext.grpo_token_loss(logp, old, ref, adv, mask, 0.8, 1.2, 0.04 if use_ref else 0.0, cispo)
        ref_loss, ref_dl = self._eager(logp, old, ref, adv, mask, 0.8, 1.2, 0.04 if use_ref else 0.0, cispo)
        torch.testing.assert_close(loss, ref_loss, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(dl, ref_dl, rtol=1e-4, atol=1e-5)


class TestAgentOnGpu:
    def test_dqn_learn_gpu(self):
        from agilerl_amd.algorithms.dqn import DQN
        from agilerl_amd.spaces import Box, Discrete

        agent = DQN(Box(-1, 1, (8,)), Discrete(4), device=DEV)
        batch = {
            "obs": torch.randn(64, 8, device=DEV),
            "action": torch.randint(0, 4, (64,), device=DEV),
            "reward": torch.randn(64, device=DEV),
            "next_obs": torch.randn(64, 8, device=DEV),
            "done": torch.zeros(64, device=DEV),
        }
        loss = agent.learn(batch)
        assert np.isfinite(loss)

    def test_ppo_cycle_gpu(self):
        from agilerl_amd.algorithms.ppo import PPO
        from agilerl_amd.components.rollout_buffer import RolloutBuffer
        from agilerl_amd.envs import LunarLanderVecEnv
        from agilerl_amd.rollouts.on_policy import collect_rollouts

        env = LunarLanderVecEnv(8, seed=0)
        agent = PPO(env.single_observation_space, env.single_action_space,
                    learn_step=16, batch_size=64, device=DEV)
        buf = RolloutBuffer(16, 8, device=DEV, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
        obs, done, _ = collect_rollouts(agent, env, buf, 16)
        stats = agent.learn(buf)
        assert np.isfinite(stats["policy_loss"])


class TestDQNGraph:
    def test_graphed_update_matches_eager(self):
        from agilerl_amd.algorithms.dqn import DQN
        from agilerl_amd.spaces import Box, Discrete

        torch.manual_seed(0)
        obs_s, act_s = Box(-1, 1, (8,)), Discrete(4)
        # identical init for both agents
        torch.manual_seed(7)
        eager = DQN(obs_s, act_s, device=DEV, lr=1e-3)
        torch.manual_seed(7)
        graphed = DQN(obs_s, act_s, device=DEV, lr=1e-3, cudagraphs=True)
        for p, q in zip(eager.actor.parameters(), graphed.actor.parameters()):
            torch.testing.assert_close(p, q)

        batches = []
        for _ in range(5):
            batches.append({
                "obs": torch.randn(64, 8, device=DEV),
                "action": torch.randint(0, 4, (64,), device=DEV),
                "reward": torch.randn(64, device=DEV),
                "next_obs": torch.randn(64, 8, device=DEV),
                "done": (torch.rand(64, device=DEV) < 0.2).float(),
            })
        for b in batches:
            l1 = eager.learn(dict(b))
            l2 = graphed.learn(dict(b))
        for p, q in zip(eager.actor.parameters(), graphed.actor.parameters()):
            torch.testing.assert_close(p, q, rtol=1e-3, atol=1e-4)
        for p, q in zip(eager.actor_target.parameters(), graphed.actor_target.parameters()):
            torch.testing.assert_close(p, q, rtol=1e-3, atol=1e-4)


@pytest.mark.gpu
def test_swiglu_matches_eager():
    import torch.nn.functional as F

    from agilerl_amd.ops.swiglu import swiglu

    torch.manual_seed(0)
    for shape in ((128, 256), (3, 17, 264), (5, 1000)):
        g = torch.randn(*shape, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        u = torch.randn(*shape, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        out = swiglu(g, u)
        ref_g = g.detach().float().requires_grad_(True)
        ref_u = u.detach().float().requires_grad_(True)
        ref = F.silu(ref_g) * ref_u
        torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)
        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.float())
        torch.testing.assert_close(g.grad.float(), ref_g.grad, atol=3e-2, rtol=3e-2)
        torch.testing.assert_close(u.grad.float(), ref_u.grad, atol=3e-2, rtol=3e-2)


@pytest.mark.gpu
def test_llama_swiglu_patch_matches_unpatched():
    from transformers import AutoConfig, AutoModelForCausalLM

    from agilerl_amd.architectures.llama_patches import patch_llama_swiglu

    cfg = AutoConfig.for_model(
        "llama", vocab_size=64, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, pad_token_id=0,
    )
    torch.manual_seed(0)
    model = AutoModelForCausalLM.from_config(cfg).to("cuda", torch.bfloat16).eval()
    ids = torch.randint(0, 64, (2, 16), device="cuda")
    with torch.no_grad():
        before = model(ids).logits.float()
    n = patch_llama_swiglu(model)
    assert n == 2
    with torch.no_grad():
        after = model(ids).logits.float()
    torch.testing.assert_close(after, before, atol=5e-2, rtol=5e-2)


@pytest.mark.gpu
def test_paged_attn_decode_matches_eager():
    """HIP paged-attention decode vs the fp32 eager reference: random
    pools, ragged lengths, GQA 4:1 and 1:1, head dims 64/128."""
    from agilerl_amd.ops.paged_attn import _eager_reference, paged_attention_decode

    torch.manual_seed(0)
    for (Hq, Hkv, D) in ((8, 2, 128), (4, 4, 64)):
        B, S, P = 3, 16, 32
        q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
        kp = torch.randn(P, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
        vp = torch.randn(P, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
        lengths = torch.tensor([37, 5, 64], device="cuda", dtype=torch.int32)
        max_pages = 5
        table = torch.randperm(P, device="cuda")[: B * max_pages].reshape(B, max_pages).to(torch.int32)
        out = paged_attention_decode(q, kp, vp, table, lengths)
        ref = _eager_reference(q.float(), kp.float(), vp.float(), table, lengths,
                               1.0 / D ** 0.5)
        torch.testing.assert_close(out, ref, atol=2e-2, rtol=2e-2)


class TestSegmentTreeKernels:
    """HIP segtree kernels vs the CPU torch tree (same API, ground truth)."""

    def _build_pair(self, capacity, n_init, seed=0):
        from agilerl_amd.components.segment_tree import MinSegmentTree, SumSegmentTree

        g = torch.Generator().manual_seed(seed)
        vals = torch.rand(n_init, generator=g) + 0.01
        idx = torch.arange(n_init)
        cpu_sum, gpu_sum = SumSegmentTree(capacity, "cpu"), SumSegmentTree(capacity, DEV)
        cpu_min, gpu_min = MinSegmentTree(capacity, "cpu"), MinSegmentTree(capacity, DEV)
        for t in (cpu_sum, cpu_min):
            t.update(idx, vals)
        for t in (gpu_sum, gpu_min):
            t.update(idx.to(DEV), vals.to(DEV))
        return cpu_sum, gpu_sum, cpu_min, gpu_min, g

    def test_update_matches_cpu(self):
        cpu_sum, gpu_sum, cpu_min, gpu_min, g = self._build_pair(1000, 1000)
        # scattered partial update including the propagation overlap case
        up_idx = torch.randint(0, 1000, (256,), generator=g)
        up_val = torch.rand(256, generator=g) + 0.01
        # dedup (duplicate leaves are nondeterministic by contract)
        up_idx = torch.unique(up_idx)
        up_val = up_val[: up_idx.numel()]
        for t, d in ((cpu_sum, "cpu"), (cpu_min, "cpu"), (gpu_sum, DEV), (gpu_min, DEV)):
            t.update(up_idx.to(d), up_val.to(d))
        torch.testing.assert_close(gpu_sum.tree.cpu(), cpu_sum.tree, rtol=1e-5, atol=1e-5)
        torch.testing.assert_close(gpu_min.tree.cpu(), cpu_min.tree, rtol=0, atol=0)
        assert float(gpu_sum.root) == pytest.approx(float(cpu_sum.root), rel=1e-5)

    def test_retrieve_matches_cpu(self):
        cpu_sum, gpu_sum, *_ , g = self._build_pair(4096, 4096, seed=1)
        total = float(cpu_sum.root)
        prefix = torch.rand(512, generator=g) * total * 0.999
        got = gpu_sum.retrieve(prefix.to(DEV)).cpu()
        want = cpu_sum.retrieve(prefix)
        torch.testing.assert_close(got, want)

    def test_retrieve_large_capacity_beyond_lds(self):
        # capacity 64k: tree = 128k nodes, far beyond the 8192-float LDS
        # stage — exercises the LDS->HBM descent transition
        cpu_sum, gpu_sum, *_ , g = self._build_pair(65536, 65536, seed=2)
        total = float(cpu_sum.root)
        prefix = torch.rand(1024, generator=g) * total * 0.999
        got = gpu_sum.retrieve(prefix.to(DEV)).cpu()
        want = cpu_sum.retrieve(prefix)
        torch.testing.assert_close(got, want)

    def test_fused_per_sample_consistent(self):
        """per_sample idx/weights == the composed eager computation given
        the same uniform jitter."""
        ext = extension()
        cpu_sum, gpu_sum, cpu_min, gpu_min, g = self._build_pair(2048, 1500, seed=3)
        B, size, beta = 256, 1500, 0.5
        rand01 = torch.rand(B, generator=g)
        idx, w = ext.per_sample(gpu_sum.tree, gpu_min.tree, rand01.to(DEV), size, beta)
        # eager reference on CPU with identical stratified prefixes
        total = float(cpu_sum.root)
        u = (torch.arange(B).float() + rand01) / B * total
        want_idx = cpu_sum.retrieve(u).clamp_(max=size - 1)
        torch.testing.assert_close(idx.cpu(), want_idx)
        p = cpu_sum.get(want_idx) / total
        p_min = float(cpu_min.min()) / total
        max_w = (p_min * size) ** (-beta)
        want_w = ((p * size).clamp(min=1e-12) ** (-beta)) / max_w
        torch.testing.assert_close(w.cpu(), want_w, rtol=1e-4, atol=1e-5)

    def test_per_buffer_end_to_end_gpu(self):
        from agilerl_amd.components import PrioritizedReplayBuffer

        buf = PrioritizedReplayBuffer(4096, device=DEV, storage_device=DEV, seed=0)
        for _ in range(8):
            buf.add(
                obs=torch.randn(64, 4, device=DEV),
                action=torch.randint(0, 2, (64,), device=DEV),
                reward=torch.randn(64, device=DEV),
                next_obs=torch.randn(64, 4, device=DEV),
                done=torch.zeros(64, device=DEV),
            )
        batch = buf.sample(128, beta=0.4)
        assert batch["idxs"].shape == (128,)
        assert batch["weights"].shape == (128,)
        assert batch["weights"].max() <= 1.0 + 1e-5
        assert (batch["idxs"] < len(buf)).all()
        # priority update round-trips through the HIP kernel
        buf.update_priorities(batch["idxs"], torch.rand(128, device=DEV) + 0.1)
        batch2 = buf.sample(128, beta=0.6)
        assert batch2["weights"].isfinite().all()


class TestPagedAttnKernel:
    """Flash-decoding paged-attention kernel vs the eager reference."""

    def _setup(self, B, Hq, Hkv, D, S, lens, seed=0):
        g = torch.Generator(device="cpu").manual_seed(seed)
        max_pages = (max(lens) + S - 1) // S
        P = B * max_pages + 4
        q = torch.randn(B, Hq, D, generator=g).bfloat16()
        k_pool = torch.randn(P, S, Hkv, D, generator=g).bfloat16()
        v_pool = torch.randn(P, S, Hkv, D, generator=g).bfloat16()
        # non-trivial page tables: shuffled page assignment
        perm = torch.randperm(P, generator=g)
        table = perm[: B * max_pages].reshape(B, max_pages).int()
        lengths = torch.tensor(lens, dtype=torch.int32)
        return q, k_pool, v_pool, table, lengths

    def _check(self, B, Hq, Hkv, D, S, lens, seed=0):
        from agilerl_amd.ops.paged_attn import _eager_reference, paged_attention_decode

        q, k, v, tbl, ln = self._setup(B, Hq, Hkv, D, S, lens, seed)
        scale = D ** -0.5
        want = _eager_reference(q.float(), k.float(), v.float(), tbl, ln, scale)
        got = paged_attention_decode(
            q.to(DEV), k.to(DEV), v.to(DEV), tbl.to(DEV), ln.to(DEV), scale)
        torch.testing.assert_close(got.cpu(), want, rtol=2e-2, atol=2e-2)

    def test_llama8b_shape_ragged(self):
        # Llama-3-8B decode shape: 32 q heads, 8 kv heads, D=128
        self._check(B=8, Hq=32, Hkv=8, D=128, S=16,
                    lens=[1021, 7, 512, 300, 64, 1, 999, 128])

    def test_long_sequence_multi_split(self):
        self._check(B=2, Hq=8, Hkv=8, D=128, S=16, lens=[4096, 3000], seed=1)

    def test_small_head_dim(self):
        self._check(B=3, Hq=4, Hkv=2, D=64, S=16, lens=[33, 17, 80], seed=2)

    def test_tiny_model_shape(self):
        self._check(B=4, Hq=2, Hkv=1, D=16, S=4, lens=[9, 3, 15, 6], seed=3)

    def test_non_pow2_fallback_shape(self):
        # D=96 falls outside the flash-decoding constraints -> basic kernel
        self._check(B=2, Hq=4, Hkv=4, D=96, S=16, lens=[40, 60], seed=4)


@pytest.mark.gpu
class TestPagedEngineGpu:
    def _model_engine(self, use_graph=True):
        from transformers import AutoConfig, AutoModelForCausalLM

        from agilerl_amd.llm.decode_engine import DecodeEngine

        cfg = AutoConfig.for_model(
            "llama", vocab_size=128, hidden_size=256, intermediate_size=512,
            num_hidden_layers=2, num_attention_heads=8, num_key_value_heads=2,
            max_position_embeddings=256, pad_token_id=0)
        torch.manual_seed(0)
        model = AutoModelForCausalLM.from_config(cfg).bfloat16().to(DEV)
        engine = DecodeEngine(model, num_pages=64, page_size=16)
        if not use_graph:
            engine._use_decode_graph = False
        return model, engine

    def test_runner_logits_match_hf_forward_bf16(self):
        """Teacher-forced: the paged runner's per-step logits must match an
        HF forward over the same prefix within bf16 tolerance (trajectory
        parity is pinned exactly on CPU fp32; bf16 argmax can legitimately
        flip near-ties, so GPU compares logits, not token ids)."""
        model, engine = self._model_engine(use_graph=False)
        torch.manual_seed(1)
        prompt = torch.randint(1, 128, (11,)).to(DEV)
        forced = torch.randint(1, 128, (6,)).tolist()
        sid = engine.submit(prompt, max_new_tokens=1)
        engine._admit()
        engine._prefill([engine.active[sid]])
        runner = engine._paged_runner
        seq = engine.active[sid]
        prefix = prompt.tolist()
        for tok in forced:
            engine.cache._ensure_capacity(sid, 1)  # page for the new token
            table = runner.build_table([sid], DEV)
            pos = torch.tensor([engine.cache.lengths[sid]], device=DEV)
            logits = runner.decode_step(
                torch.tensor([tok], device=DEV), pos, table)
            engine.cache.lengths[sid] += 1
            prefix.append(tok)
            full = torch.tensor([prefix], device=DEV)
            ref = model(input_ids=full).logits[0, -1]
            torch.testing.assert_close(
                logits[0].float(), ref.float(), rtol=5e-2, atol=5e-1)

    def test_graphed_engine_matches_ungraphed_engine(self):
        """hipGraph-captured decode must be step-for-step identical to the
        same paged path run eagerly (same kernels, same order)."""
        torch.manual_seed(5)
        prompts = [torch.randint(1, 128, (n,)) for n in (12, 5, 20, 3)]
        outs = []
        for use_graph in (True, False):
            _, engine = self._model_engine(use_graph=use_graph)
            if use_graph:
                assert engine._use_decode_graph
            sids = [engine.submit(p.to(DEV), max_new_tokens=8) for p in prompts]
            results = engine.run_all()
            outs.append([results[s].cpu() for s in sids])
        for a, b in zip(outs[0], outs[1]):
            torch.testing.assert_close(a, b)

    def test_graphed_engine_continuous_batching(self):
        """Staggered admission/retirement with the graph decoder: zombie
        slots, slot reuse, and mid-flight admission all produce the same
        sequences as the ungraphed engine."""
        torch.manual_seed(9)
        prompts = [torch.randint(1, 128, (n,)) for n in (7, 15, 4, 10, 6)]
        lens = [5, 12, 3, 9, 7]
        outs = []
        for use_graph in (True, False):
            _, engine = self._model_engine(use_graph=use_graph)
            engine.max_batch = 2  # forces waiting queue + slot churn
            sids = [engine.submit(p.to(DEV), max_new_tokens=n)
                    for p, n in zip(prompts, lens)]
            results = engine.run_all()
            outs.append([results[s].cpu() for s in sids])
        for a, b in zip(outs[0], outs[1]):
            torch.testing.assert_close(a, b)


class TestKernelEdgeCases:
    """non-divisible vocab, huge row counts,
    empty/zero masks, non-contiguous inputs, bf16 GAE paths."""

    def test_row_lse_gather_non_divisible_vocab(self):
        ext = extension()
        for V in (63, 1001, 32013, 127999):
            rows = 17
            logits = torch.randn(rows, V, device=DEV)
            targets = torch.randint(0, V, (rows,), device=DEV)
            lp, lse = ext.row_lse_gather(logits, targets, 1.0)
            ref_lse = torch.logsumexp(logits, dim=-1)
            ref_lp = logits.gather(1, targets.unsqueeze(1)).squeeze(1) - ref_lse
            torch.testing.assert_close(lse, ref_lse, rtol=1e-4, atol=1e-4,
                                       msg=lambda m: f"V={V}: {m}")
            torch.testing.assert_close(lp, ref_lp, rtol=1e-4, atol=1e-4)

    def test_fused_logprobs_many_rows(self):
        # >64k rows exercises the chunk loop and grid clamps
        from agilerl_amd.ops.fused_logprobs import fused_linear_logprobs

        N, H, V = 70_000, 64, 1024
        h = torch.randn(N, H, device=DEV)
        w = torch.randn(V, H, device=DEV)
        t = torch.randint(0, V, (N,), device=DEV)
        out = fused_linear_logprobs(h, w, t)
        assert out.shape == (N,)
        # spot-check a slice against eager
        sl = slice(65_530, 65_600)
        logits = h[sl] @ w.t()
        ref = logits.gather(1, t[sl].unsqueeze(1)).squeeze(1) - torch.logsumexp(logits, -1)
        torch.testing.assert_close(out[sl], ref, rtol=1e-3, atol=1e-3)

    def test_grpo_loss_empty_mask(self):
        from agilerl_amd.ops.grpo_loss import grpo_policy_loss

        B, T = 4, 16
        logp = (torch.randn(B, T, device=DEV) * 0.1).requires_grad_(True)
        old = logp.detach() + 0.05
        adv = torch.randn(B, T, device=DEV)
        mask = torch.zeros(B, T, device=DEV)  # nothing unmasked
        loss = grpo_policy_loss(logp, old, adv, mask)
        assert torch.isfinite(loss)
        assert float(loss) == 0.0
        loss.backward()
        assert torch.isfinite(logp.grad).all()
        assert float(logp.grad.abs().sum()) == 0.0

    def test_gae_scan_non_contiguous_and_bf16(self):
        T, N = 32, 65
        base = torch.randn(T, N * 2, device=DEV)
        rewards = base[:, ::2]  # non-contiguous view
        assert not rewards.is_contiguous()
        values = torch.randn(T, N, device=DEV)
        dones = (torch.rand(T, N, device=DEV) < 0.1).float()
        last_value = torch.randn(N, device=DEV)
        adv, ret = ops.gae_scan(rewards, values, dones, last_value, 0.99, 0.95)
        adv_ref, _ = ops.gae_scan(rewards.cpu(), values.cpu(), dones.cpu(),
                                  last_value.cpu(), 0.99, 0.95)
        torch.testing.assert_close(adv.cpu(), adv_ref, rtol=1e-4, atol=1e-5)
        # bf16 rollout tensors: dispatcher casts to fp32 for the scan
        adv_bf, ret_bf = ops.gae_scan(
            rewards.bfloat16(), values.bfloat16(), dones.bfloat16(),
            last_value.bfloat16(), 0.99, 0.95)
        assert adv_bf.dtype == torch.float32
        torch.testing.assert_close(adv_bf.cpu(), adv_ref, rtol=2e-2, atol=2e-2)

    def test_c51_non_contiguous_rewards(self):
        B, A = 33, 51
        dist = torch.softmax(torch.randn(B, A, device=DEV), -1)
        rewards2 = torch.randn(B, 2, device=DEV)
        rewards = rewards2[:, 0]  # stride-2 view
        dones = torch.zeros(B, device=DEV)
        support = torch.linspace(-10, 10, A, device=DEV)
        out = ops.c51_project(dist, rewards, dones, support, 0.99, -10, 10)
        ref = ops.c51_project(dist.cpu(), rewards.cpu(), dones.cpu(),
                              support.cpu(), 0.99, -10, 10)
        torch.testing.assert_close(out.cpu(), ref, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(out.sum(-1).cpu(), torch.ones(B), rtol=1e-4, atol=1e-4)

    def test_group_advantage_single_group_and_group_of_one(self):
        r = torch.randn(8, device=DEV)
        # one big group
        out = ops.group_advantage(r, 8)
        ref = ops.group_advantage(r.cpu(), 8)
        torch.testing.assert_close(out.cpu(), ref, rtol=1e-4, atol=1e-5)
        # degenerate group_size=1: zero advantages, no NaN from std
        out1 = ops.group_advantage(r, 1)
        assert torch.isfinite(out1).all()


class TestSkinnyGemm:
    """Decode-shape GEMM kernel vs torch matmul."""

    @pytest.mark.parametrize("M,N,K", [
        (1, 4096, 4096), (8, 1024, 4096), (8, 4096, 4096),
        (16, 14336, 4096), (8, 4096, 14336), (3, 129, 64),
    ])
    def test_matches_matmul(self, M, N, K):
        ext = extension()
        g = torch.Generator().manual_seed(M + N + K)
        x = (torch.randn(M, K, generator=g) * 0.2).bfloat16().to(DEV)
        w = (torch.randn(N, K, generator=g) * 0.2).bfloat16().to(DEV)
        got = ext.skinny_gemm(x, w)
        want = (x.float() @ w.float().t())
        torch.testing.assert_close(got.float(), want, rtol=2e-2, atol=2e-1)

    def test_decode_runner_uses_it(self):
        """_fast_linear must route through the kernel (same numerics as the
        module) for bf16 M<=16 bias-free projections."""
        import torch.nn as nn

        from agilerl_amd.llm.paged_llama import _fast_linear

        lin = nn.Linear(4096, 1024, bias=False).bfloat16().to(DEV)
        x = torch.randn(8, 4096).bfloat16().to(DEV)
        out = _fast_linear(lin, x)
        ref = lin(x)
        torch.testing.assert_close(out.float(), ref.float(), rtol=2e-2, atol=2e-1)

    def test_lora_delta_preserved(self):
        import torch.nn as nn

        from agilerl_amd.llm.lora import LoraConfig, LoraLinear
        from agilerl_amd.llm.paged_llama import _fast_linear

        base = nn.Linear(512, 256, bias=False).bfloat16().to(DEV)
        mod = LoraLinear(base, LoraConfig(r=8, lora_alpha=16), adapters=("self",))
        mod = mod.to(DEV).bfloat16()
        with torch.no_grad():
            mod.lora_B["self"].normal_(0, 0.05)  # nonzero delta
        mod.active_adapter = "self"
        x = torch.randn(4, 512).bfloat16().to(DEV)
        out = _fast_linear(mod, x)
        ref = mod(x)
        torch.testing.assert_close(out.float(), ref.float(), rtol=3e-2, atol=3e-1)
        assert not torch.allclose(out.float(), base(x).float(), rtol=1e-3, atol=1e-3)


class TestKernelSanitizer:
    """SURVEY 5.2 analog of compute-sanitizer CI: re-run the kernel
    numerics suite with serialized kernel launches (AMD_SERIALIZE_KERNEL=3
    forces a sync after every launch, so async faults surface at the
    guilty kernel instead of a later sync point)."""

    def test_kernels_under_serialized_launches(self):
        import os
        import subprocess
        import sys

        env = dict(os.environ)
        env["AMD_SERIALIZE_KERNEL"] = "3"
        targets = [
            "tests/test_ops_gpu.py::TestGaeScan",
            "tests/test_ops_gpu.py::TestNStep",
            "tests/test_ops_gpu.py::TestC51",
            "tests/test_ops_gpu.py::TestGroupAdvantage",
            "tests/test_ops_gpu.py::TestSegmentTreeKernels",
            "tests/test_ops_gpu.py::TestSkinnyGemm",
            "tests/test_ops_gpu.py::TestPagedAttnKernel",
        ]
        out = subprocess.run(
            [sys.executable, "-m", "pytest", *targets, "-q", "-m", "gpu",
             "-p", "no:cacheprovider"],
            capture_output=True, text=True, timeout=600, env=env,
        )
        assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-1000:]
