"""First-party space primitive tests."""

import numpy as np

from agilerl_amd.spaces import (
    Box,
    DictSpace,
    Discrete,
    MultiBinary,
    MultiDiscrete,
    TupleSpace,
    flatdim,
    is_image_space,
    space_shape,
)


class TestSpaces:
    def test_box(self):
        b = Box(-1.0, 1.0, (3,))
        s = b.sample()
        assert s.shape == (3,) and b.contains(s)
        assert not b.contains(np.array([2.0, 0, 0]))
        assert b == Box(-1.0, 1.0, (3,))
        assert b != Box(-2.0, 2.0, (3,))

    def test_box_infinite_bounds(self):
        b = Box(-np.inf, np.inf, (2,))
        s = b.sample()
        assert np.isfinite(s).all()

    def test_discrete(self):
        d = Discrete(5)
        assert 0 <= d.sample() < 5
        assert d.contains(4) and not d.contains(5)
        assert space_shape(d) == (5,)  # one-hot width

    def test_multidiscrete(self):
        md = MultiDiscrete([3, 4, 2])
        s = md.sample()
        assert s.shape == (3,) and md.contains(s)
        assert flatdim(md) == 9

    def test_multibinary(self):
        mb = MultiBinary(4)
        s = mb.sample()
        assert set(np.unique(s)).issubset({0, 1})
        assert mb.contains(s)

    def test_dict_space(self):
        ds = DictSpace({"a": Box(-1, 1, (2,)), "b": Discrete(3)})
        s = ds.sample()
        assert ds.contains(s)
        assert flatdim(ds) == 5
        assert list(ds.keys()) == ["a", "b"]

    def test_tuple_space(self):
        ts = TupleSpace([Box(-1, 1, (2,)), Discrete(3)])
        s = ts.sample()
        assert ts.contains(s)
        assert flatdim(ts) == 5

    def test_image_predicate(self):
        assert is_image_space(Box(0, 255, (3, 84, 84)))
        assert not is_image_space(Box(-1, 1, (8,)))

    def test_seeding(self):
        a, b = Box(-1, 1, (4,), seed=3), Box(-1, 1, (4,), seed=3)
        np.testing.assert_array_equal(a.sample(), b.sample())


class TestUtilsExtras:
    def test_sampling_utils(self):
        import torch

        from agilerl_amd.utils.sampling_utils import process_logits, sample_from_logits

        logits = torch.randn(2, 10)
        out = process_logits(logits, temperature=0.5, top_k=3)
        assert torch.isinf(out).sum() >= 2 * 7  # 7 masked per row
        s = sample_from_logits(logits, top_p=0.9)
        assert s.shape == (2, 1)

    def test_kv_cache(self):
        import torch

        from agilerl_amd.utils.cache import Cache

        c = Cache()
        k = torch.randn(2, 4, 3, 8)
        v = torch.randn(2, 4, 3, 8)
        ck, cv = c.update(0, k, v)
        assert ck.shape[2] == 3
        ck, cv = c.update(0, k[:, :, :1], v[:, :, :1])
        assert ck.shape[2] == 4 and c.length == 4
        c.trim(2)
        assert c.length == 2

    def test_chat_template_fallback(self):
        from agilerl_amd.llm.chat import apply_chat_template

        text = apply_chat_template(None, "hi", system_prompt="sys")
        assert "hi" in text and "sys" in text

    def test_pz_auto_reset_wrapper(self):
        from agilerl_amd.wrappers.pettingzoo_wrappers import AutoResetParallelWrapper
        from tests.test_vector import _ToyPZEnv

        env = AutoResetParallelWrapper(_ToyPZEnv())
        obs, _ = env.reset()
        for _ in range(3):
            obs, r, te, tr, info = env.step({"a0": 1, "a1": 0})
        assert "final_observation" in info
        assert obs["a0"][0] == 0.0  # reset happened
