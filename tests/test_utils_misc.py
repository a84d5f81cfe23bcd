"""KV-cache and sampling-utility tests (legacy offline-RL support stack)."""

import torch

from agilerl_amd.utils.cache import Cache
from agilerl_amd.utils.sampling_utils import (
    process_logits,
    sample_from_logits,
    top_k_logits,
    top_p_logits,
)


class TestCache:
    def test_update_concatenates_per_layer(self):
        c = Cache()
        k1, v1 = torch.randn(2, 2, 4, 8), torch.randn(2, 2, 4, 8)  # (B,H,T,D)
        k, v = c.update(0, k1, v1)
        assert k.shape[2] == 4
        k, v = c.update(0, torch.randn(2, 2, 3, 8), torch.randn(2, 2, 3, 8))
        assert k.shape[2] == 7 and c.length == 7
        # independent layers
        k, _ = c.update(1, k1, v1)
        assert k.shape[2] == 4

    def test_trim_and_reset(self):
        c = Cache()
        c.update(0, torch.randn(1, 2, 10, 4), torch.randn(1, 2, 10, 4))
        c.trim(6)
        k, _ = c.update(0, torch.randn(1, 2, 1, 4), torch.randn(1, 2, 1, 4))
        assert k.shape[2] == 7
        c.reset()
        k, _ = c.update(0, torch.randn(1, 2, 2, 4), torch.randn(1, 2, 2, 4))
        assert k.shape[2] == 2 and c.length == 2


class TestSampling:
    def test_top_k_masks_all_but_k(self):
        logits = torch.tensor([[1.0, 5.0, 3.0, 2.0]])
        out = top_k_logits(logits, 2)
        assert torch.isfinite(out[0, 1]) and torch.isfinite(out[0, 2])
        assert out[0, 0] == -float("inf") or out[0, 0] < -1e9
        assert out[0, 3] == -float("inf") or out[0, 3] < -1e9

    def test_top_p_keeps_minimal_nucleus(self):
        logits = torch.log(torch.tensor([[0.5, 0.3, 0.15, 0.05]]))
        out = top_p_logits(logits, 0.7)
        probs = torch.softmax(out, dim=-1)
        assert probs[0, 3] < 1e-6  # tail dropped
        assert probs[0, 0] > 0.5

    def test_sample_respects_mask(self):
        torch.manual_seed(0)
        logits = torch.zeros(64, 4)
        logits[:, 2] = 100.0
        samples = sample_from_logits(process_logits(logits, temperature=1.0, top_k=1))
        assert (samples == 2).all()
