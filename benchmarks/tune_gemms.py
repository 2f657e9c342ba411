#!/usr/bin/env python3
"""Offline hipBLASLt solution search for the Llama train-step GEMM shapes.

    python benchmarks/tune_gemms.py [--m 32768] [--cap 0] [--iters 5]
        [--shapes qkv,o,gate_up,down]

For each shape, times hipBLASLt solutions (`ops._mfx_gemm.search`) on the
real bf16 TN problem and compares the winner against the library
heuristic (what torch.matmul uses) and against torch.nn.functional.linear
itself. Winners go to stdout as JSON — round 2 pins them via
`_mfx_gemm.run(x, w, index)`.

The step profile (profiles/llama8b_1gpu_r01_final2_kernel_stats.txt) has
hipBLASLt at ~58% of step time, so a few percent here is a few percent
end-to-end.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

# llama3-8b, B=8 S=4096 -> M = 32768 (fwd); K/N per projection
SHAPES = {
    "qkv": (4096, 6144),        # K=h, N=(nq+2nkv)*hd
    "o": (4096, 4096),
    "gate_up": (4096, 28672),   # 2*intermediate
    "down": (14336, 4096),
    "lm_head": (4096, 128256),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=32768)
    p.add_argument("--cap", type=int, default=0,
                   help="max solutions to time per shape (0 = all)")
    p.add_argument("--iters", type=int, default=5)
    p.add_argument("--shapes", type=str, default="qkv,o,gate_up,down")
    args = p.parse_args()

    import torch

    from metaflow_amd.ops import _mfx_gemm as G

    assert torch.cuda.is_available(), "tuner needs a GPU"
    dev = torch.device("cuda", 0)
    results = {}
    for name in args.shapes.split(","):
        K, N = SHAPES[name]
        torch.manual_seed(0)
        x = torch.randn(args.m, K, dtype=torch.bfloat16, device=dev) * 0.1
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1

        # correctness first: heuristic path vs torch
        ref = torch.nn.functional.linear(x, w)
        got = G.heuristic(x, w)
        err = (got.float() - ref.float()).abs().max().item()
        denom = ref.float().abs().max().item()
        assert err / denom < 2e-2, "heuristic mismatch %g" % (err / denom)

        # time torch's own path
        for _ in range(2):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(args.iters):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        torch_ms = (time.time() - t0) / args.iters * 1000

        idxs, ms = G.search(x, w, args.iters, args.cap)
        best_idx = int(idxs[0])
        best_ms = float(ms[0])

        # verify the winner's numerics before trusting it
        got2 = G.run(x, w, best_idx)
        err2 = (got2.float() - ref.float()).abs().max().item()
        assert err2 / denom < 2e-2, "winner mismatch %g" % (err2 / denom)

        tflops = 2.0 * args.m * K * N / 1e12
        results[name] = {
            "m": args.m, "k": K, "n": N,
            "torch_ms": round(torch_ms, 3),
            "best_ms": round(best_ms, 3),
            "best_index": best_idx,
            "speedup_vs_torch": round(torch_ms / best_ms, 3),
            "best_tflops": round(tflops / (best_ms / 1e3), 1),
            "torch_tflops": round(tflops / (torch_ms / 1e3), 1),
            "solutions_timed": int(idxs.numel()),
        }
        print(name, json.dumps(results[name]), flush=True)
    print(json.dumps({"tuned": results}), flush=True)


if __name__ == "__main__":
    main()
