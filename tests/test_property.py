"""Property-based tests (hypothesis) for core numerics: segment trees,
GAE/n-step scans, C51 projection, sequence packing."""

import numpy as np
import pytest
import torch

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from agilerl_amd import ops
from agilerl_amd.components.segment_tree import MinSegmentTree, SumSegmentTree
from agilerl_amd.llm.packing import pack_padded_batch, unpack_values

COMMON = dict(deadline=None, max_examples=25)


@settings(**COMMON)
@given(
    n=st.integers(2, 64),
    seed=st.integers(0, 10_000),
)
def test_sum_tree_prefix_retrieve_matches_cumsum(n, seed):
    rng = np.random.default_rng(seed)
    vals = rng.uniform(0.01, 5.0, n).astype(np.float32)
    tree = SumSegmentTree(n)
    tree.update(torch.arange(n), torch.from_numpy(vals))
    total = float(tree.sum())
    assert total == pytest.approx(vals.sum(), rel=1e-4)
    prefix = torch.from_numpy(rng.uniform(0, vals.sum() * 0.999, 8).astype(np.float32))
    idx = tree.retrieve(prefix).numpy()
    cum = np.concatenate([[0.0], np.cumsum(vals)])
    for p, i in zip(prefix.numpy(), idx):
        assert cum[i] <= p + 1e-3 and p < cum[i + 1] + 1e-3


@settings(**COMMON)
@given(n=st.integers(2, 64), seed=st.integers(0, 10_000))
def test_min_tree_matches_numpy_min(n, seed):
    rng = np.random.default_rng(seed)
    vals = rng.uniform(0.01, 5.0, n).astype(np.float32)
    tree = MinSegmentTree(n)
    tree.update(torch.arange(n), torch.from_numpy(vals))
    assert float(tree.min()) == pytest.approx(vals.min(), rel=1e-5)


@settings(**COMMON)
@given(
    t=st.integers(1, 24),
    n=st.integers(1, 6),
    gamma=st.floats(0.8, 0.999),
    lam=st.floats(0.8, 1.0),
    seed=st.integers(0, 10_000),
)
def test_gae_scan_matches_naive_recursion(t, n, gamma, lam, seed):
    g = torch.Generator().manual_seed(seed)
    rewards = torch.randn(t, n, generator=g)
    values = torch.randn(t, n, generator=g)
    dones = (torch.rand(t, n, generator=g) < 0.2).float()
    last_value = torch.randn(n, generator=g)
    adv, ret = ops.gae_scan(rewards, values, dones, last_value, gamma, lam)
    # naive per-env python recursion (dones[i] = done-after-step-i cuts both
    # the bootstrap and the lambda carry of step i)
    for env in range(n):
        next_adv, next_val = 0.0, float(last_value[env])
        expect = np.zeros(t)
        for i in range(t - 1, -1, -1):
            nd = 1.0 - float(dones[i, env])
            delta = float(rewards[i, env]) + gamma * next_val * nd - float(values[i, env])
            next_adv = delta + gamma * lam * nd * next_adv
            expect[i] = next_adv
            next_val = float(values[i, env])
        np.testing.assert_allclose(adv[:, env].numpy(), expect, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(ret, adv + values)


@settings(**COMMON)
@given(
    b=st.integers(1, 6),
    t=st.integers(2, 20),
    seed=st.integers(0, 10_000),
)
def test_packing_roundtrip_any_mask(b, t, seed):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, 100, (b, t), generator=g)
    # random mask but every row keeps at least one token, left-aligned
    lengths = torch.randint(1, t + 1, (b,), generator=g)
    mask = (torch.arange(t).unsqueeze(0) < lengths.unsqueeze(1)).long()
    pack = pack_padded_batch(ids, mask)
    n_real = int(mask.sum())
    assert pack["packed_ids"].shape == (1, n_real)
    assert int(pack["cu_seqlens"][-1]) == n_real
    # position ids restart per sequence
    cu = pack["cu_seqlens"]
    for i in range(b):
        seg = pack["position_ids"][0, cu[i]:cu[i + 1]]
        torch.testing.assert_close(seg, torch.arange(int(lengths[i])))
    # values written back land in original positions
    vals = torch.arange(n_real, dtype=torch.float32)
    unpacked = unpack_values(vals, pack, fill=-1.0)
    assert unpacked.shape == (b, t)
    assert (unpacked.reshape(-1)[pack["indices"]] == vals).all()
    assert float(unpacked.reshape(-1).sum()) == pytest.approx(
        vals.sum() - (b * t - n_real), rel=1e-5
    )


@settings(**COMMON)
@given(
    batch=st.integers(1, 8),
    atoms=st.integers(11, 51),
    gamma=st.floats(0.9, 0.999),
    seed=st.integers(0, 10_000),
)
def test_c51_projection_is_distribution(batch, atoms, gamma, seed):
    g = torch.Generator().manual_seed(seed)
    v_min, v_max = -10.0, 10.0
    probs = torch.softmax(torch.randn(batch, atoms, generator=g), dim=-1)
    rewards = torch.randn(batch, generator=g) * 5
    dones = (torch.rand(batch, generator=g) < 0.3).float()
    support = torch.linspace(v_min, v_max, atoms)
    out = ops.c51_project(probs, rewards, dones, support, gamma, v_min, v_max)
    assert out.shape == (batch, atoms)
    assert (out >= -1e-6).all()
    torch.testing.assert_close(out.sum(-1), torch.ones(batch), atol=1e-4, rtol=1e-4)


@settings(**COMMON)
@given(seed=st.integers(0, 10_000))
def test_space_samples_in_bounds(seed):
    from agilerl_amd.spaces import (Box, Discrete, MultiBinary, MultiDiscrete,
                                    DictSpace, TupleSpace, flatdim)

    rng = np.random.default_rng(seed)
    lo, hi = sorted(rng.uniform(-5, 5, 2))
    spaces = [
        Box(lo, hi + 1e-3, (3,)),
        Discrete(int(rng.integers(2, 10))),
        MultiDiscrete([2, 3, 4]),
        MultiBinary(5),
    ]
    for sp in spaces:
        s = sp.sample()
        if isinstance(sp, Box):
            assert (np.asarray(s) >= sp.low - 1e-6).all()
            assert (np.asarray(s) <= sp.high + 1e-6).all()
        elif isinstance(sp, Discrete):
            assert 0 <= s < sp.n
        elif isinstance(sp, MultiDiscrete):
            assert all(0 <= v < n for v, n in zip(s, sp.nvec))
        else:
            assert set(np.asarray(s).ravel()) <= {0, 1}
        assert flatdim(sp) > 0
    d = DictSpace({"a": spaces[0], "b": spaces[1]})
    t = TupleSpace((spaces[0], spaces[3]))
    assert flatdim(d) == flatdim(spaces[0]) + flatdim(spaces[1])
    assert set(d.sample()) == {"a", "b"}
    assert len(t.sample()) == 2
