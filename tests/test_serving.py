"""Continuous-batching serving engine: token-exactness vs per-request
generate(), slot reuse, varlen decode (CPU; the GPU kernel variant is
covered by test_ops_gpu + the serving example on-box)."""

import pytest
import torch

from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
from metaflow_amd.serving import ContinuousBatcher


@pytest.fixture(scope="module")
def tiny_model():
    torch.manual_seed(0)
    cfg = LlamaConfig.tiny(vocab=128, seq=256)
    return LlamaForCausalLM(cfg).eval()


def test_continuous_batching_token_exact(tiny_model):
    """Concurrent requests with different prompt lengths and budgets
    produce exactly the tokens sequential generate() produces."""
    torch.manual_seed(1)
    prompts = [
        ([5, 9, 17, 4], 6),
        (list(range(2, 30)), 5),
        ([100, 101], 8),
        ([7] * 11, 4),
        ([64, 3, 99, 12, 54, 23], 7),
    ]
    batcher = ContinuousBatcher(tiny_model, max_batch=2, max_len=128)
    reqs = [batcher.submit(p, n) for p, n in prompts]
    out = batcher.run()
    assert set(out) == {r.id for r in reqs}
    for req, (prompt, n) in zip(reqs, prompts):
        ref = tiny_model.generate(
            torch.tensor([prompt]), n)[0, len(prompt):].tolist()
        assert out[req.id] == ref, (req.id, out[req.id], ref)
        assert req.done


def test_batcher_slot_reuse(tiny_model):
    """More requests than slots: slots are recycled and every request
    completes with its full token budget."""
    batcher = ContinuousBatcher(tiny_model, max_batch=2, max_len=64)
    reqs = [batcher.submit([3 + i, 7, 11], 3 + (i % 3))
            for i in range(7)]
    out = batcher.run()
    for i, r in enumerate(reqs):
        assert len(out[r.id]) == 3 + (i % 3)


def test_varlen_decode_cpu_matches_per_slot():
    """attn_decode_varlen CPU path: per-slot lengths, inactive slot
    yields zeros."""
    from metaflow_amd.ops import kernels as K

    torch.manual_seed(0)
    B, H, Hkv, Lmax = 3, 4, 2, 50
    q = torch.randn(B, H, 1, 128, dtype=torch.bfloat16)
    kc = torch.randn(B, Hkv, Lmax, 128, dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    lengths = [10, 0, 37]
    o = K.attn_decode_varlen(q, kc, vc, lengths, 0.0883)
    assert torch.all(o[1] == 0)
    for b in (0, 2):
        ref = K.attention_ref(q[b:b + 1], kc[b:b + 1, :, :lengths[b]],
                              vc[b:b + 1, :, :lengths[b]], 0.0883,
                              causal=False)
        err = ((o[b:b + 1] - ref).float().norm()
               / (ref.float().norm() + 1e-8)).item()
        assert err < 2e-2


def test_serving_flow_example(tmp_path):
    """examples/serving_flow.py end to end: a 2-rank gang drains the
    request stream through ContinuousBatcher (gloo/CPU here; the same
    flow runs GPU-resident on a box)."""
    import os
    import subprocess
    import sys

    from .test_runtime import REPO

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MFX_NUM_GPUS"] = "0"
    ds = tmp_path / "ds"
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples",
                                      "serving_flow.py"),
         "--datastore-root", str(ds), "run",
         "--ranks", "2", "--num-requests", "6"],
        env=env, capture_output=True, text=True, timeout=600)
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert "served" in proc.stdout


def test_continuous_batching_mixtral():
    """The serving engine drives the MoE family too: greedy tokens from
    the batcher match per-request Mixtral generate() (eager decode —
    MoE routing is shape-dynamic, so no graph capture)."""
    from metaflow_amd.models.mixtral import (
        MixtralConfig,
        MixtralForCausalLM,
    )

    torch.manual_seed(0)
    m = MixtralForCausalLM(MixtralConfig.tiny(vocab=128, seq=256)).eval()
    prompts = [([5, 9, 17, 4], 5), (list(range(2, 20)), 4),
               ([100, 101], 6)]
    batcher = ContinuousBatcher(m, max_batch=2, max_len=64)
    reqs = [batcher.submit(p, n) for p, n in prompts]
    out = batcher.run()
    for req, (prompt, n) in zip(reqs, prompts):
        ref = m.generate(torch.tensor([prompt]),
                         n)[0, len(prompt):].tolist()
        assert out[req.id] == ref, (req.id, out[req.id], ref)


def test_chunked_prefill_token_exact(tiny_model):
    """prefill_chunk: chunked admission produces exactly the tokens the
    whole-prompt batcher produces (chunk boundary mid-prompt, chunk
    bigger than a short prompt, several requests racing slots)."""
    prompts = [
        (list(range(2, 25)), 5),      # 23 tokens -> 3 chunks of 8
        ([5, 9, 17], 6),              # shorter than one chunk
        ([7] * 16, 4),                # exact chunk multiple
        ([100, 101, 102, 103, 104], 7),
    ]
    ref_b = ContinuousBatcher(tiny_model, max_batch=2, max_len=128)
    ref_reqs = [ref_b.submit(p, n) for p, n in prompts]
    ref_out = ref_b.run()
    chk_b = ContinuousBatcher(tiny_model, max_batch=2, max_len=128,
                              prefill_chunk=8)
    chk_reqs = [chk_b.submit(p, n) for p, n in prompts]
    chk_out = chk_b.run()
    for rr, cr in zip(ref_reqs, chk_reqs):
        assert chk_out[cr.id] == ref_out[rr.id], (chk_out[cr.id],
                                                  ref_out[rr.id])
        assert cr.done


def test_chunked_prefill_bounds_step_latency(tiny_model):
    """A long prompt never runs >prefill_chunk prompt tokens in one
    step(): active decoders are not stalled behind a monolithic
    prefill."""
    seen = []
    orig_forward = tiny_model.forward

    def spy(tokens, *a, **kw):
        seen.append(tokens.shape[1])
        return orig_forward(tokens, *a, **kw)

    tiny_model.forward = spy
    try:
        b = ContinuousBatcher(tiny_model, max_batch=2, max_len=256,
                              prefill_chunk=16)
        b.submit(list(range(2, 2 + 100)), 3)   # 100-token prompt
        b.submit([5, 6, 7], 3)
        b.run()
    finally:
        tiny_model.forward = orig_forward
    assert max(seen) <= 16, seen
