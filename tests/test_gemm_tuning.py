"""Tuned-GEMM plumbing: table parsing and CPU fallback (the GPU replay
path is exercised by benchmarks/tune_gemms.py on-box)."""

import json

import torch


def test_tuned_linear_cpu_fallback(tmp_path, monkeypatch):
    from metaflow_amd.ops import gemm

    f = tmp_path / "tune.json"
    f.write_text(json.dumps({"64,32,16": 123}))
    monkeypatch.setenv("MFX_GEMM_TUNE_FILE", str(f))
    gemm.reset_tune_table()
    x = torch.randn(64, 32)
    w = torch.randn(16, 32)
    out = gemm.tuned_linear(x, w)   # CPU -> plain F.linear
    ref = torch.nn.functional.linear(x, w)
    assert torch.equal(out, ref)
    assert gemm._load_table() == {(64, 32, 16): 123}
    gemm.reset_tune_table()


def test_tuned_linear_no_table():
    from metaflow_amd.ops import gemm

    gemm.reset_tune_table()
    x = torch.randn(4, 8, dtype=torch.bfloat16)
    w = torch.randn(6, 8, dtype=torch.bfloat16)
    out = gemm.tuned_linear(x, w)
    assert out.shape == (4, 6)
