#!/usr/bin/env python3
"""A/B the tuned-GEMM path against plain F.linear, per shape, fwd+bwd.

Times the REAL autograd composition (F.linear forward + its backward)
vs TunedLinear with pinned indices — unlike tune_gemms.py's per-GEMM
torch baseline, this includes no artificial transpose copies, so it
answers 'does pinning help end-to-end'.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

SHAPES = {
    "qkv": (4096, 6144),
    "o": (4096, 4096),
    "gate_up": (4096, 28672),
    "down": (14336, 4096),
    "lm_head": (4096, 128256),
}


def main():
    import torch

    from metaflow_amd.ops import gemm

    table = gemm._load_table()
    print("table entries:", len(table), flush=True)
    dev = "cuda"
    M = 32768
    for name, (K, N) in SHAPES.items():
        torch.manual_seed(0)
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
        dy = torch.randn(M, N, dtype=torch.bfloat16, device=dev) * 0.1
        x.requires_grad_(True)
        w.requires_grad_(True)

        def timeit(f, n=10):
            for _ in range(3):
                f()
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(n):
                f()
            torch.cuda.synchronize()
            return (time.time() - t0) / n * 1000

        def torch_step():
            x.grad = None
            w.grad = None
            out = torch.nn.functional.linear(x, w)
            out.backward(dy)

        def tuned_step():
            x.grad = None
            w.grad = None
            out = gemm.tuned_linear(x, w)
            out.backward(dy)

        tm = timeit(torch_step)
        tu = timeit(tuned_step)
        # numerics check
        x.grad = None; w.grad = None
        ref = torch.nn.functional.linear(x, w); ref.backward(dy)
        gx, gw = x.grad.clone(), w.grad.clone()
        x.grad = None; w.grad = None
        out = gemm.tuned_linear(x, w); out.backward(dy)
        ef = (out - ref).float().norm() / ref.float().norm()
        ex = (x.grad - gx).float().norm() / gx.float().norm()
        ew = (w.grad - gw).float().norm() / gw.float().norm()
        print("%-8s torch %7.3f ms  tuned %7.3f ms  speedup %.3f  "
              "err f/x/w %.1e %.1e %.1e"
              % (name, tm, tu, tm / tu, ef, ex, ew), flush=True)
        del x, w, dy


if __name__ == "__main__":
    main()
