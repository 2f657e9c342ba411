"""Greedy speculative decoding: token-exact vs plain target-greedy
generate() across k values, draft quality, and prompt shapes (CPU;
the acceptance rule guarantees exactness, these tests enforce it)."""

import pytest
import torch

from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
from metaflow_amd.speculative import speculative_generate


@pytest.fixture(scope="module")
def models():
    torch.manual_seed(0)
    target = LlamaForCausalLM(LlamaConfig.tiny(vocab=128, seq=256)).eval()
    torch.manual_seed(1)
    draft = LlamaForCausalLM(LlamaConfig.tiny(vocab=128, seq=256)).eval()
    return target, draft


@pytest.mark.parametrize("k", [1, 2, 4, 7])
def test_speculative_token_exact(models, k):
    target, draft = models
    prompt = torch.tensor([[5, 9, 17, 4, 33, 2, 77]])
    ref = target.generate(prompt, 24)
    out, stats = speculative_generate(target, draft, prompt, 24, k=k)
    assert torch.equal(out, ref), (k, out.tolist(), ref.tolist())
    assert out.shape[1] == prompt.shape[1] + 24
    assert stats["target_steps"] <= 1 + 24   # never worse than greedy


def test_speculative_selfdraft_accepts_everything(models):
    """Draft == target: every proposal accepted, target steps shrink
    by ~k per round."""
    target, _ = models
    prompt = torch.tensor([[11, 3, 8]])
    ref = target.generate(prompt, 20)
    out, stats = speculative_generate(target, target, prompt, 20, k=4)
    assert torch.equal(out, ref)
    assert stats["accepted"] == stats["proposed"]
    # 1 prefill + ceil(19/5) verify rounds (each round nets k+1 tokens)
    assert stats["target_steps"] <= 6


def test_speculative_smaller_draft(models):
    """Draft with a different architecture (fewer layers/heads) — the
    realistic deployment shape — still token-exact."""
    target, _ = models
    torch.manual_seed(7)
    cfg = LlamaConfig.tiny(vocab=128, seq=256)
    cfg.num_layers = max(1, cfg.num_layers // 2)
    draft = LlamaForCausalLM(cfg).eval()
    prompt = torch.tensor([[9, 1, 2, 3, 4]])
    ref = target.generate(prompt, 16)
    out, _ = speculative_generate(target, draft, prompt, 16, k=3)
    assert torch.equal(out, ref)


def test_speculative_short_budget(models):
    target, draft = models
    prompt = torch.tensor([[42, 43]])
    for n in (1, 2, 3):
        ref = target.generate(prompt, n)
        out, _ = speculative_generate(target, draft, prompt, n, k=4)
        assert torch.equal(out, ref), n


def test_speculative_edge_shapes(models):
    """Prompt of length 1, and k far exceeding the budget."""
    target, draft = models
    for prompt, n, k in [([7], 8, 4), ([3, 5], 4, 16), ([2] * 30, 6, 5)]:
        pt = torch.tensor([prompt])
        ref = target.generate(pt, n)
        out, stats = speculative_generate(target, draft, pt, n, k=k)
        assert torch.equal(out, ref), (prompt, n, k)
        assert len(out[0]) == len(prompt) + n
