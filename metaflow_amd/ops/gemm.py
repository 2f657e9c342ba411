"""Tuned hipBLASLt GEMMs: replay offline-searched solution indices.

`benchmarks/tune_gemms.py` sweeps every hipBLASLt solution for a shape
(`_mfx_gemm.search`, see ops/csrc/gemm_lt.hip) and the winners get
pinned here: set ``MFX_GEMM_TUNE_FILE`` to a JSON of
``{"M,K,N": solution_index, ...}`` and `tuned_linear` routes matching
shapes through ``_mfx_gemm.run`` — everything else (and every CPU call)
falls back to ``torch.nn.functional.linear``. First sweep found +7.6%
on the gate_up projection (profiles/bench_results_r01.md).
"""

import json
import os

import torch

_table = None
_ext = None


def _load_table():
    global _table
    if _table is None:
        _table = {}
        path = os.environ.get("MFX_GEMM_TUNE_FILE")
        if path and os.path.isfile(path):
            with open(path) as f:
                raw = json.load(f)
            for key, idx in raw.items():
                m, k, n = (int(v) for v in key.split(","))
                _table[(m, k, n)] = int(idx)
    return _table


def _gemm_ext():
    global _ext
    if _ext is None:
        from . import _mfx_gemm

        _ext = _mfx_gemm
    return _ext


def reset_tune_table():
    """Re-read MFX_GEMM_TUNE_FILE on next use (tests/tuning loops)."""
    global _table
    _table = None


def tuned_linear(x, weight):
    """F.linear with pinned hipBLASLt solutions where tuned.

    x: [..., K] bf16, weight: [N, K] bf16. Only exact (M, K, N) matches
    use a pinned index — autograd-safe because tuning applies to the
    forward GEMM only (wrap in autograd.Function when the backward
    shapes get tuned too; round-2 item).
    """
    if not (x.is_cuda and x.dtype == torch.bfloat16):
        return torch.nn.functional.linear(x, weight)
    table = _load_table()
    if not table:
        return torch.nn.functional.linear(x, weight)
    K = x.shape[-1]
    M = x.numel() // K
    N = weight.shape[0]
    idx = table.get((M, K, N))
    if idx is None or x.requires_grad or weight.requires_grad:
        return torch.nn.functional.linear(x, weight)
    out = _gemm_ext().run(x.reshape(M, K), weight, idx)
    return out.reshape(*x.shape[:-1], N)
