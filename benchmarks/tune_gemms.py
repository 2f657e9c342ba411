#!/usr/bin/env python3
"""Offline hipBLASLt solution search for the Llama train-step GEMM shapes.

    python benchmarks/tune_gemms.py [--m 32768] [--cap 0] [--iters 3]
        [--shapes qkv,o,gate_up,down,lm_head] [--modes fwd,dx,dw]
        [--out metaflow_amd/ops/gemm_table.json]

For each (shape, mode), times hipBLASLt solutions (`ops._mfx_gemm.search`)
on the real bf16 problem — mode fwd is the TN forward, dx the NN
input-grad, dw the NT weight-grad (see ops/csrc/gemm_lt.hip) — then
re-times the top 8 with more iters, verifies the winner's numerics
against torch, and writes the winners table consumed by ops/gemm.py.

The round-1 step profile has hipBLASLt at ~58% of step time at ~49% of
its peak, so a few percent here is a few percent end-to-end.
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
MODE_ID = {"fwd": 0, "dx": 1, "dw": 2}


def _operands(mode, M, K, N, dev):
    import torch

    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
    dy = torch.randn(M, N, dtype=torch.bfloat16, device=dev) * 0.1
    if mode == "fwd":
        return w, x, lambda: torch.nn.functional.linear(x, w)
    if mode == "dx":
        return w, dy, lambda: dy @ w
    return x, dy, lambda: dy.t().contiguous() @ x   # dw


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=32768)
    p.add_argument("--cap", type=int, default=0,
                   help="max solutions to time per problem (0 = all)")
    p.add_argument("--iters", type=int, default=3)
    p.add_argument("--refine-iters", type=int, default=10)
    p.add_argument("--shapes", type=str,
                   default="qkv,o,gate_up,down,lm_head")
    p.add_argument("--modes", type=str, default="fwd,dx,dw")
    p.add_argument("--out", type=str, default=None,
                   help="write winners table JSON here")
    args = p.parse_args()

    import torch

    from metaflow_amd.ops import _mfx_gemm as G

    assert torch.cuda.is_available(), "tuner needs a GPU"
    dev = torch.device("cuda", 0)
    results = {}
    table = {}
    for name in args.shapes.split(","):
        K, N = SHAPES[name]
        M = args.m
        for mode in args.modes.split(","):
            mid = MODE_ID[mode]
            a, b, torch_fn = _operands(mode, M, K, N, dev)

            # correctness of the layout derivation first
            ref = torch_fn().float()
            got = G.heuristic(mid, a, b).float()
            denom = ref.abs().max().item()
            err = (got - ref).abs().max().item() / denom
            assert err < 2e-2, "%s/%s heuristic mismatch %g" % (name, mode,
                                                                err)

            # torch's own path timing
            for _ in range(2):
                torch_fn()
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(args.iters):
                torch_fn()
            torch.cuda.synchronize()
            torch_ms = (time.time() - t0) / args.iters * 1000

            idxs, ms = G.search(mid, a, b, args.iters, args.cap)
            assert idxs.numel() > 0, "no solutions for %s/%s" % (name,
                                                                 mode)
            # re-time the top 8 with more iters (first pass is noisy)
            top = [(int(idxs[i]), float(ms[i]))
                   for i in range(min(8, idxs.numel()))]
            refined = [(i, G.time_one(mid, a, b, i, args.refine_iters))
                       for i, _ in top]
            refined.sort(key=lambda t: t[1])
            best_idx, best_ms = refined[0]

            got2 = G.run(mid, a, b, best_idx).float()
            err2 = (got2 - ref).abs().max().item() / denom
            assert err2 < 2e-2, "winner mismatch %g" % err2

            tflops = 2.0 * M * K * N / 1e12
            key = "%s|%d,%d,%d" % (mode, M, K, N)
            table[key] = best_idx
            results["%s/%s" % (name, mode)] = {
                "m": M, "k": K, "n": N,
                "torch_ms": round(torch_ms, 3),
                "best_ms": round(best_ms, 3),
                "best_index": best_idx,
                "speedup_vs_torch": round(torch_ms / best_ms, 3),
                "best_tflops": round(tflops / (best_ms / 1e3), 1),
                "torch_tflops": round(tflops / (torch_ms / 1e3), 1),
                "solutions_timed": int(idxs.numel()),
            }
            print(name, mode, json.dumps(results["%s/%s" % (name, mode)]),
                  flush=True)
    if args.out:
        with open(args.out, "w") as f:
            json.dump(table, f, indent=1, sort_keys=True)
        print("wrote", args.out, flush=True)
    print(json.dumps({"tuned": results}), flush=True)


if __name__ == "__main__":
    main()
