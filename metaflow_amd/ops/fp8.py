"""FP8 (OCP E4M3) training linear with delayed scaling — gfx950-native.

gfx950's matrix cores run E4M3 at 2x the bf16 rate; hipBLASLt exposes it
via HIP_R_8F_E4M3 (ops/csrc/gemm_lt.hip: gemm_lt_fp8, fp32 accumulate,
descale folded into alpha). This module implements the standard
delayed-scaling recipe (per-tensor scale from an amax HISTORY so the
cast is a single fused pass without a pre-scan of the current tensor):

  * forward GEMM in E4M3 (x and w quantized with their delayed scales);
  * backward GEMMs in bf16 (dgrad/wgrad keep bf16 accuracy — the
    standard first rung of fp8 training);
  * amax history window (default 16 steps) updated per forward.

``Fp8Linear`` is a drop-in for models.llama.Linear; CPU (and missing
extension) falls back to bf16 F.linear so the module stays testable
everywhere. E4M3 max normal = 448.
"""

import torch

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


def _gemm_ext():
    from . import _mfx_gemm

    return _mfx_gemm


def quantize_e4m3(t, scale):
    """bf16/fp32 -> float8_e4m3fn storage viewed as uint8 (saturating).

    GPU bf16 goes through the one-pass quant_e4m3 HIP kernel (native
    v_cvt_pk_fp8_f32); everything else uses the torch composition."""
    if t.is_cuda and t.dtype == torch.bfloat16 and t.numel() % 8 == 0:
        from .kernels import hip_ext

        return hip_ext().quant_e4m3(t.contiguous(), float(scale))
    x = (t.float() * scale).clamp(-E4M3_MAX, E4M3_MAX)
    return x.to(torch.float8_e4m3fn).view(torch.uint8)


def quantize_e5m2(t, scale):
    """bf16/fp32 -> float8_e5m2 storage viewed as uint8 (saturating).

    E5M2's wider exponent range suits GRADIENT tensors (dy spans more
    orders of magnitude than activations). GPU bf16 goes through the
    one-pass quant_e5m2 HIP kernel (v_cvt_pk_bf8_f32) — written at
    round-2 end, compile-checked; the kernel's first on-HW numerics
    run is a round-3 task before switching the dy path to e5m2."""
    if t.is_cuda and t.dtype == torch.bfloat16 and t.numel() % 8 == 0:
        from .kernels import hip_ext

        return hip_ext().quant_e5m2(t.contiguous(), float(scale))
    x = (t.float() * scale).clamp(-E5M2_MAX, E5M2_MAX)
    return x.to(torch.float8_e5m2).view(torch.uint8)


class _Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, scale_x, scale_w, mod):
        K = x.shape[-1]
        M = x.numel() // K
        x8 = quantize_e4m3(x.reshape(M, K), scale_x)
        w8 = quantize_e4m3(w, scale_w)
        out = _gemm_ext().fp8(x8, w8, 1.0 / (scale_x * scale_w))
        ctx.mod = mod
        if mod is not None and mod.fp8_bwd:
            # fp8 backward: keep the QUANTIZED operands (half the
            # activation memory of saving bf16 x) — dgrad/wgrad rerun
            # the same TN-form fp8 GEMM on byte-transposed views
            ctx.save_for_backward(x8, w8)
            ctx.meta = (scale_x, scale_w, x.shape, K, M)
        else:
            ctx.save_for_backward(x, w)
        return out.reshape(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        mod = ctx.mod
        if mod is None or not mod.fp8_bwd:
            x, w = ctx.saved_tensors
            K = x.shape[-1]
            M = x.numel() // K
            dy2 = dy.reshape(M, -1)
            dx = (dy2 @ w).reshape(x.shape)
            dw = dy2.t() @ x.reshape(M, K)
            return dx, dw, None, None, None
        x8, w8 = ctx.saved_tensors
        sx, sw, xshape, K, M = ctx.meta
        dy2 = dy.reshape(M, -1).contiguous()
        # dy scale is DELAYED from the previous backwards' amax history
        # (first fp8 backward seeds it from this dy: one host sync once)
        ady = float(mod.amax_dy.max())
        if ady <= 0:
            ady = float(dy2.detach().abs().max())
        sdy = E4M3_MAX / ady if ady > 0 else 1.0
        dy8 = quantize_e4m3(dy2, sdy)
        # dgrad: dx[M,K] = dy[M,N] @ W[N,K]   = fp8(dy8, w8^T)
        dx = _gemm_ext().fp8(dy8, w8.t().contiguous(), 1.0 / (sdy * sw))
        # wgrad: dw[N,K] = dy^T[N,M] @ x[M,K] = fp8(dy8^T, x8^T)
        dw = _gemm_ext().fp8(dy8.t().contiguous(), x8.t().contiguous(),
                             1.0 / (sdy * sx))
        with torch.no_grad():
            mod.amax_dy[mod._bstep % mod.history] = \
                dy2.detach().abs().max().float()
        mod._bstep += 1
        return dx.reshape(xshape), dw, None, None, None


def _in_recompute():
    """True inside torch.utils.checkpoint's backward recompute — state
    updates (amax history, step counter) must NOT re-run there, and the
    recompute must reuse the ORIGINAL forward's scales or the
    recomputed activations diverge from what autograd saved."""
    try:
        from torch.utils.checkpoint import is_recomputing

        return bool(is_recomputing())
    except Exception:
        pass
    try:
        # this torch predates is_recomputing(): checkpoint recompute is
        # the only way this module's forward runs INSIDE a backward
        # graph task
        return torch._C._current_graph_task_id() != -1
    except Exception:
        return False


class Fp8Linear(torch.nn.Module):
    """Bias-free linear with an E4M3 forward GEMM and delayed scaling."""

    def __init__(self, din, dout, dtype=torch.bfloat16, history=16,
                 fp8_bwd=False):
        super().__init__()
        self.weight = torch.nn.Parameter(
            torch.empty(dout, din, dtype=dtype))
        self.history = history
        #: fp8_bwd=True additionally runs dgrad AND wgrad in E4M3
        #: (delayed-scaled dy) — the full-fp8 rung; measure loss
        #: quality before production use
        self.fp8_bwd = fp8_bwd
        self.register_buffer(
            "amax_x", torch.zeros(history), persistent=False)
        self.register_buffer(
            "amax_w", torch.zeros(history), persistent=False)
        self.register_buffer(
            "amax_dy", torch.zeros(history), persistent=False)
        self._step = 0
        self._bstep = 0

    def _scales(self):
        ax = float(self.amax_x.max())
        aw = float(self.amax_w.max())
        sx = E4M3_MAX / ax if ax > 0 else 1.0
        sw = E4M3_MAX / aw if aw > 0 else 1.0
        return sx, sw

    def _update_amax(self, x):
        i = self._step % self.history
        with torch.no_grad():
            self.amax_x[i] = x.detach().abs().max().float()
            self.amax_w[i] = self.weight.detach().abs().max().float()
        self._step += 1

    def forward(self, x):
        if not (x.is_cuda and hasattr(torch, "float8_e4m3fn")):
            return torch.nn.functional.linear(x, self.weight)
        if _in_recompute():
            # activation-checkpoint recompute: reuse the original
            # forward's scales, touch no state
            sx, sw = self._last_scales
        else:
            sx, sw = self._scales()  # delayed: prior steps' amax window
            self._last_scales = (sx, sw)
            self._update_amax(x)
        if self._step == 1:
            # no history yet: first step runs bf16 (standard warmup)
            return torch.nn.functional.linear(x, self.weight)
        return _Fp8LinearFn.apply(x, self.weight, sx, sw, self)
