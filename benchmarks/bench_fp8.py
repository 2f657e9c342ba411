#!/usr/bin/env python3
"""FP8 E4M3 GEMM measurement on the train-step shapes (gfx950 native,
2x bf16 matrix-core rate; hipBLASLt HIP_R_8F_E4M3 path).

Reports TFLOP/s per shape vs the bf16 heuristic, verifies numerics
against the bf16 product (fp8 tolerance), and times Fp8Linear
(delayed-scaling module) fwd+bwd vs bf16 F.linear.
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


def timeit(f, n=10):
    import torch

    for _ in range(3):
        f()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1000


def main():
    import torch

    from metaflow_amd.ops import _mfx_gemm as G
    from metaflow_amd.ops.fp8 import E4M3_MAX, Fp8Linear, quantize_e4m3

    dev = "cuda"
    M = 32768
    for name, (K, N) in SHAPES.items():
        torch.manual_seed(0)
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.1
        sx = E4M3_MAX / x.abs().max().float().item()
        sw = E4M3_MAX / w.abs().max().float().item()
        x8 = quantize_e4m3(x, sx)
        w8 = quantize_e4m3(w, sw)
        ref = torch.nn.functional.linear(x, w).float()
        got = G.fp8(x8, w8, 1.0 / (sx * sw)).float()
        rel = ((got - ref).norm() / ref.norm()).item()
        assert rel < 6e-2, (name, rel)
        bf16_ms = timeit(lambda: torch.nn.functional.linear(x, w))
        fp8_ms = timeit(lambda: G.fp8(x8, w8, 1.0 / (sx * sw)))
        tf = 2.0 * M * K * N / 1e12
        print("%-8s bf16 %7.3f ms (%6.0f TF/s)   fp8 %7.3f ms "
              "(%6.0f TF/s)  speedup %.2f  rel_err %.3f"
              % (name, bf16_ms, tf / (bf16_ms / 1e3),
                 fp8_ms, tf / (fp8_ms / 1e3), bf16_ms / fp8_ms, rel),
              flush=True)

    # module-level: delayed-scaling linear fwd+bwd vs bf16
    K, N = SHAPES["gate_up"]
    lin = Fp8Linear(K, N).to(dev)
    torch.nn.init.normal_(lin.weight, std=0.02)
    x = (torch.randn(M, K, dtype=torch.bfloat16, device=dev)
         * 0.1).detach().requires_grad_(True)
    lin(x)  # warmup step fills amax history

    def fp8_step():
        x.grad = None
        out = lin(x)
        out.backward(torch.ones_like(out) * 1e-3)

    wb = torch.nn.Parameter(lin.weight.detach().clone())

    def bf16_step():
        x.grad = None
        out = torch.nn.functional.linear(x, wb)
        out.backward(torch.ones_like(out) * 1e-3)

    print("Fp8Linear gate_up fwd+bwd: fp8 %.3f ms vs bf16 %.3f ms"
          % (timeit(fp8_step), timeit(bf16_step)), flush=True)


if __name__ == "__main__":
    main()
