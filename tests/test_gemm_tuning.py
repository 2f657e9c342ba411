"""Tuned-GEMM plumbing: table parsing and CPU fallback (the GPU replay
path is exercised by benchmarks/tune_gemms.py + test_ops_gpu on-box)."""

import json

import torch


def test_tuned_linear_cpu_fallback(tmp_path, monkeypatch):
    from metaflow_amd.ops import gemm

    f = tmp_path / "tune.json"
    f.write_text(json.dumps({"fwd|64,32,16": 123, "dx|64,32,16": 4,
                             "dw|64,32,16": 5}))
    monkeypatch.setenv("MFX_GEMM_TUNE_FILE", str(f))
    gemm.reset_tune_table()
    x = torch.randn(64, 32)
    w = torch.randn(16, 32)
    out = gemm.tuned_linear(x, w)   # CPU -> plain F.linear
    ref = torch.nn.functional.linear(x, w)
    assert torch.equal(out, ref)
    assert gemm._load_table() == {("fwd", 64, 32, 16): 123,
                                  ("dx", 64, 32, 16): 4,
                                  ("dw", 64, 32, 16): 5}
    gemm.reset_tune_table()


def test_tune_table_v1_format(tmp_path, monkeypatch):
    """Round-1 fwd-only keys ("M,K,N") still parse."""
    from metaflow_amd.ops import gemm

    f = tmp_path / "tune.json"
    f.write_text(json.dumps({"64,32,16": 7}))
    monkeypatch.setenv("MFX_GEMM_TUNE_FILE", str(f))
    gemm.reset_tune_table()
    assert gemm._load_table() == {("fwd", 64, 32, 16): 7}
    gemm.reset_tune_table()


def test_tuned_linear_no_table(monkeypatch):
    from metaflow_amd.ops import gemm

    monkeypatch.setenv("MFX_GEMM_TUNE_FILE", "/nonexistent")
    gemm.reset_tune_table()
    x = torch.randn(4, 8, dtype=torch.bfloat16)
    w = torch.randn(6, 8, dtype=torch.bfloat16)
    out = gemm.tuned_linear(x, w)
    assert out.shape == (4, 6)
    gemm.reset_tune_table()


def test_shipped_table_parses():
    """The in-repo gemm_table.json (if present) must parse into
    (mode, M, K, N) -> index entries."""
    import os

    from metaflow_amd.ops import gemm

    gemm.reset_tune_table()
    if os.path.isfile(gemm._DEFAULT_TABLE):
        table = gemm._load_table()
        assert table, "shipped table exists but parsed empty"
        for (mode, m, k, n), idx in table.items():
            assert mode in ("fwd", "dx", "dw")
            assert m > 0 and k > 0 and n > 0 and idx >= 0
    gemm.reset_tune_table()


def test_quantize_e5m2_cpu_reference():
    """quantize_e5m2's CPU composition: saturating scale->e5m2 cast
    matches a manual reference; values round-trip within e5m2's 2-bit
    mantissa; saturation clamps at 57344."""
    import torch

    from metaflow_amd.ops.fp8 import E5M2_MAX, quantize_e5m2

    torch.manual_seed(0)
    x = torch.randn(64, 8, dtype=torch.bfloat16) * 1e-3
    scale = E5M2_MAX / x.abs().max().float().item()
    q = quantize_e5m2(x, scale)
    assert q.dtype == torch.uint8 and q.shape == x.shape
    back = q.view(torch.float8_e5m2).float() / scale
    rel = (back - x.float()).abs() / x.float().abs().clamp_min(1e-12)
    assert rel.median() < 0.15          # 2 mantissa bits ~= 12.5% ulp
    # saturation
    big = torch.full((8,), 1e6, dtype=torch.bfloat16)
    qb = quantize_e5m2(big, 1.0).view(torch.float8_e5m2).float()
    assert (qb == E5M2_MAX).all()
