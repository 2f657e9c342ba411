#!/usr/bin/env python3
"""Standalone attention-kernel timings at the flagship shape
(B=8, H=32, Hkv=8, S=4096, D=128 — llama3-8b at batch 8 x seq 4096).

Times attn_fwd and attn_bwd (and the bwd parts via CUDA events around
the composite call), reports ms/launch and effective TFLOP/s so kernel
work can be judged without a full bench run.
"""

import argparse
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--b", type=int, default=8)
    p.add_argument("--h", type=int, default=32)
    p.add_argument("--hkv", type=int, default=8)
    p.add_argument("--s", type=int, default=4096)
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()

    import torch

    from metaflow_amd.ops.kernels import hip_ext

    ext = hip_ext()
    dev = "cuda"
    B, H, Hkv, S, D = args.b, args.h, args.hkv, args.s, 128
    scale = 1.0 / math.sqrt(D)
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev) * 0.3
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev) * 0.3
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev) * 0.3
    o, lse = ext.attn_fwd(q, k, v, scale, True)
    dout = torch.randn_like(o)

    def timeit(f, n):
        for _ in range(3):
            f()
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.time() - t0) / n * 1000

    # causal: S^2/2 positions; fwd = 2 GEMMs, bwd = 5 (dq: S+dP+dQ? the
    # composite has delta + dkdv(4 matmuls incl recomputed S/dP) + dq(3))
    pos = B * H * (S * S / 2) * D
    fwd_ms = timeit(lambda: ext.attn_fwd(q, k, v, scale, True),
                    args.iters)
    bwd_ms = timeit(lambda: ext.attn_bwd(q, k, v, o, dout, lse, scale,
                                         True), args.iters)
    fwd_tf = 2 * 2 * pos / (fwd_ms / 1e3) / 1e12
    bwd_tf = 7 * 2 * pos / (bwd_ms / 1e3) / 1e12
    print("fwd  %.2f ms  (%.0f TF/s of 2-GEMM flops)" % (fwd_ms, fwd_tf))
    print("bwd  %.2f ms  (%.0f TF/s of 7-GEMM flops)" % (bwd_ms, bwd_tf))


if __name__ == "__main__":
    main()
