"""Python surface for the gfx950 kernel library.

Every op has two implementations:
  * the HIP kernel (used on GPU — REQUIRED there: if the extension is
    missing on a GPU box we raise KernelExtensionMissing rather than
    silently fall back to eager, so GPU runs always exercise native code);
  * a plain PyTorch fp32 reference (`*_ref`), used on CPU and by the
    numerics tests as ground truth.
"""

import math

import torch

from ..exceptions import KernelExtensionMissing

_ext = None
_ext_err = None


def hip_ext():
    """The _mfx_hip extension; raises loudly if unavailable on GPU."""
    global _ext, _ext_err
    if _ext is None and _ext_err is None:
        try:
            from . import _mfx_hip

            _ext = _mfx_hip
        except Exception as e:  # noqa: BLE001
            _ext_err = e
    if _ext is None:
        raise KernelExtensionMissing(
            "metaflow_amd._mfx_hip is not built (%s). Run "
            "`python setup.py build_ext --inplace` (hipcc, gfx950)."
            % _ext_err)
    return _ext


def extension_loaded():
    try:
        hip_ext()
        return True
    except KernelExtensionMissing:
        return False


# ============================== RMSNorm ====================================
class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        y, inv = hip_ext().rmsnorm_fwd(x, w, eps)
        ctx.save_for_backward(x, w, inv)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, inv = ctx.saved_tensors
        dx, dw = hip_ext().rmsnorm_bwd(x, w, dy.contiguous(), inv, None)
        return dx, dw.to(w.dtype), None


def rmsnorm(x, w, eps=1e-5):
    if x.is_cuda:
        return _RMSNorm.apply(x.contiguous(), w.contiguous(), eps)
    return rmsnorm_ref(x, w, eps).to(x.dtype)


class _AddRMSNorm(torch.autograd.Function):
    """(s, y) = (x + res, rmsnorm(x + res) * w) in one kernel; backward
    folds the gradient arriving at s into the norm backward (one launch)."""

    @staticmethod
    def forward(ctx, x, res, w, eps):
        s, y, inv = hip_ext().add_rmsnorm_fwd(x, res, w, eps)
        ctx.save_for_backward(s, w, inv)
        return s, y

    @staticmethod
    def backward(ctx, ds, dy):
        s, w, inv = ctx.saved_tensors
        if ds is not None:
            ds = ds.contiguous()
        dx, dw = hip_ext().rmsnorm_bwd(s, w, dy.contiguous(), inv, ds)
        # d/dx and d/dres are identical (s = x + res)
        return dx, dx, dw.to(w.dtype), None


def add_rmsnorm(x, res, w, eps=1e-5):
    """Fused residual add + RMSNorm: returns (sum, normed). res may be
    None (falls back to plain rmsnorm, sum = x)."""
    if res is None:
        return x, rmsnorm(x, w, eps)
    if x.is_cuda:
        return _AddRMSNorm.apply(x.contiguous(), res.contiguous(),
                                 w.contiguous(), eps)
    s = x + res
    return s, rmsnorm_ref(s, w, eps).to(x.dtype)


def rmsnorm_ref(x, w, eps=1e-5):
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * w.float()).to(x.dtype)


# ================================ RoPE =====================================
def rope_tables(seqlen, dim, theta=500000.0, device=None):
    """Precomputed fp32 cos/sin tables [seqlen, dim/2] (guide App B:
    on-device trig turns a memory-bound op VALU-bound). device=None keeps
    the active default device (so models built under `with torch.device`
    get device-resident tables — the kernel must never see host pointers)."""
    half = dim // 2
    freqs = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64)
                             / half))
    pos = torch.arange(seqlen, dtype=torch.float64)
    ang = torch.outer(pos, freqs)
    cos_t = ang.cos().float().contiguous()
    sin_t = ang.sin().float().contiguous()
    if device is not None:
        cos_t = cos_t.to(device)
        sin_t = sin_t.to(device)
    return cos_t, sin_t


class _Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos_t, sin_t, rows_per_pos, seqlen, pos0):
        ctx.tables = (cos_t, sin_t)
        ctx.meta = (rows_per_pos, seqlen, pos0)
        return hip_ext().rope(x, cos_t, sin_t, rows_per_pos, seqlen, pos0,
                              False)

    @staticmethod
    def backward(ctx, dy):
        cos_t, sin_t = ctx.tables
        rows_per_pos, seqlen, pos0 = ctx.meta
        dx = hip_ext().rope(dy.contiguous(), cos_t, sin_t, rows_per_pos,
                            seqlen, pos0, True)
        return dx, None, None, None, None, None


def rope(x, cos_t, sin_t, pos0=0, layout="bshd"):
    """Half-rotation RoPE. layout 'bshd': x [B,S,Hh,D] (positions on dim
    1); layout 'bhsd': x [B,Hh,S,D] (positions on dim 2 — what the
    attention kernel consumes, so fused-QKV slices need no extra copy)."""
    if layout == "bshd":
        B, S, Hh, D = x.shape
        rows_per_pos = Hh
    else:
        B, Hh, S, D = x.shape
        rows_per_pos = 1
    if x.is_cuda:
        return _Rope.apply(x.contiguous(), cos_t, sin_t, rows_per_pos, S,
                           pos0)
    return rope_ref(x, cos_t, sin_t, pos0, layout)


def rope_ref(x, cos_t, sin_t, pos0=0, layout="bshd"):
    D = x.shape[-1]
    half = D // 2
    if layout == "bshd":
        S = x.shape[1]
        c = cos_t[pos0:pos0 + S].view(1, S, 1, half).to(torch.float32)
        s = sin_t[pos0:pos0 + S].view(1, S, 1, half).to(torch.float32)
    else:
        S = x.shape[2]
        c = cos_t[pos0:pos0 + S].view(1, 1, S, half).to(torch.float32)
        s = sin_t[pos0:pos0 + S].view(1, 1, S, half).to(torch.float32)
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], -1).to(x.dtype)


# =============================== SwiGLU ====================================
class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g, u):
        ctx.save_for_backward(g, u)
        return hip_ext().swiglu_fwd(g, u)

    @staticmethod
    def backward(ctx, dy):
        g, u = ctx.saved_tensors
        dg, du = hip_ext().swiglu_bwd(g, u, dy.contiguous())
        return dg, du


def swiglu(g, u):
    if g.is_cuda:
        return _SwiGLU.apply(g.contiguous(), u.contiguous())
    return swiglu_ref(g, u)


def swiglu_ref(g, u):
    gf = g.float()
    return (torch.nn.functional.silu(gf) * u.float()).to(g.dtype)


class _SwiGLUFused(torch.autograd.Function):
    """SwiGLU over the fused [.., 2I] gate|up GEMM output — no slice
    copies; backward emits d_gateup in the same fused layout."""

    @staticmethod
    def forward(ctx, gu):
        ctx.save_for_backward(gu)
        return hip_ext().swiglu_gu_fwd(gu)

    @staticmethod
    def backward(ctx, dy):
        (gu,) = ctx.saved_tensors
        return hip_ext().swiglu_gu_bwd(gu, dy)


def swiglu_fused(gu):
    if gu.is_cuda:
        return _SwiGLUFused.apply(gu.contiguous())
    I = gu.shape[-1] // 2
    return swiglu_ref(gu[..., :I], gu[..., I:])


# ============================ cross entropy ================================
class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        loss, lse = hip_ext().cross_entropy_fwd(logits, targets,
                                                ignore_index)
        ctx.save_for_backward(logits, targets, lse)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        dlogits = hip_ext().cross_entropy_bwd(
            logits, targets, lse, dloss.contiguous().float(),
            ctx.ignore_index)
        return dlogits, None, None


def cross_entropy(logits, targets, ignore_index=-100):
    """Per-token loss [N] fp32 from bf16 logits [N, V]."""
    if logits.is_cuda:
        return _CrossEntropy.apply(logits.contiguous(),
                                   targets.contiguous(), ignore_index)
    return cross_entropy_ref(logits, targets, ignore_index)


def cross_entropy_ref(logits, targets, ignore_index=-100):
    return torch.nn.functional.cross_entropy(
        logits.float(), targets, ignore_index=ignore_index,
        reduction="none")


# ============================ fused qkv rope ===============================
class _RopeQKV(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, cos_t, sin_t, nq, nkv):
        q, k, v = hip_ext().rope_qkv_fwd(qkv, cos_t, sin_t, nq, nkv)
        ctx.save_for_backward(cos_t, sin_t)
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        cos_t, sin_t = ctx.saved_tensors
        dqkv = hip_ext().rope_qkv_bwd(dq.contiguous(), dk.contiguous(),
                                      dv.contiguous(), cos_t, sin_t)
        return dqkv, None, None, None, None


def rope_qkv(qkv, cos_t, sin_t, nq, nkv):
    """Fused QKV split + bhsd transpose + RoPE.

    qkv: the fused projection output [B, S, (nq+2*nkv)*128]; returns
    rope'd q [B,nq,S,128], k [B,nkv,S,128] and plain v [B,nkv,S,128] in
    one pass (elementwise.hip: rope_qkv_fwd_kernel) — replaces three
    transpose-contiguous copies and two rope launches per decoder layer.
    """
    if qkv.is_cuda:
        return _RopeQKV.apply(qkv.contiguous(), cos_t, sin_t, nq, nkv)
    B, S, _ = qkv.shape
    q, k, v = qkv.split([nq * 128, nkv * 128, nkv * 128], dim=-1)
    q = q.reshape(B, S, nq, 128).transpose(1, 2).contiguous()
    k = k.reshape(B, S, nkv, 128).transpose(1, 2).contiguous()
    v = v.reshape(B, S, nkv, 128).transpose(1, 2).contiguous()
    q = rope(q, cos_t, sin_t, layout="bhsd")
    k = rope(k, cos_t, sin_t, layout="bhsd")
    return q, k, v


# ============================== attention ==================================
_ATTN_ALIGN = 256  # v2 kernel tile: S must be a multiple of 256


def _pad_seq(t, s_pad):
    """Zero-pad [B,H,S,D] along dim 2 to s_pad rows."""
    B, H, S, D = t.shape
    if S == s_pad:
        return t.contiguous()
    out = t.new_zeros((B, H, s_pad, D))
    out[:, :, :S] = t
    return out


def _attn_pad_len(S, causal):
    """Padded seqlen for the kernel, or S unchanged if already aligned.

    Zero END-padding is exact for causal attention: every padded key sits
    at a position >= the real seqlen, so no real query row can attend it;
    padded query rows produce garbage that the wrappers slice off (their
    lse stays finite — zero scores — so 0*exp never makes a NaN in bwd).
    Non-causal attention WOULD attend padded keys, so there we keep the
    alignment requirement and fail loudly.
    """
    if S % _ATTN_ALIGN == 0:
        return S
    if not causal:
        raise ValueError(
            "non-causal attention requires seqlen %% %d == 0 (got %d); "
            "pad the inputs (zero-padding keys is only exact for causal)"
            % (_ATTN_ALIGN, S))
    return (S + _ATTN_ALIGN - 1) // _ATTN_ALIGN * _ATTN_ALIGN


class _Attention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, causal):
        S = q.size(2)
        Sp = _attn_pad_len(S, causal)
        qp = _pad_seq(q, Sp)
        kp = _pad_seq(k, Sp)
        vp = _pad_seq(v, Sp)
        o, lse = hip_ext().attn_fwd(qp, kp, vp, scale, causal)
        ctx.save_for_backward(qp, kp, vp, o, lse)
        ctx.scale = scale
        ctx.causal = causal
        ctx.real_s = S
        return o[:, :, :S].contiguous() if Sp != S else o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        S, Sp = ctx.real_s, q.size(2)
        dp = _pad_seq(dout, Sp)
        dq, dk, dv = hip_ext().attn_bwd(q, k, v, o, dp, lse,
                                        ctx.scale, ctx.causal)
        if Sp != S:
            dq = dq[:, :, :S].contiguous()
            dk = dk[:, :, :S].contiguous()
            dv = dv[:, :, :S].contiguous()
        return dq, dk, dv, None, None


def attention(q, k, v, scale=None, causal=True):
    """GQA flash attention. q: [B,H,S,128], k/v: [B,Hkv,S,128]."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    if q.is_cuda:
        return _Attention.apply(q.contiguous(), k.contiguous(),
                                v.contiguous(), scale, causal)
    return attention_ref(q, k, v, scale, causal)


def _ref_scores(q, k, scale, causal):
    """Scaled (and causally masked) fp32 score matrix with GQA expand."""
    G = q.size(1) // k.size(1)
    kf = k.float().repeat_interleave(G, dim=1)
    s = torch.matmul(q.float(), kf.transpose(-1, -2)) * scale
    if causal:
        S, Skv = q.size(2), k.size(2)
        mask = torch.ones(S, Skv, dtype=torch.bool, device=q.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    return s


def attention_ref(q, k, v, scale=None, causal=True):
    """fp32 eager reference (GQA)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    G = q.size(1) // k.size(1)
    s = _ref_scores(q, k, scale, causal)
    p = torch.softmax(s, dim=-1)
    vf = v.float().repeat_interleave(G, dim=1)
    return torch.matmul(p, vf).to(q.dtype)


def attn_fwd_raw(q, k, v, scale, causal=True):
    """Non-autograd attention forward returning (o, lse).

    ``lse[b,h,s] = log sum_kv exp(score*scale)`` — the merge statistic for
    context-parallel (ring) attention. HIP kernel on GPU; fp32 reference
    on CPU.
    """
    if q.is_cuda:
        S = q.size(2)
        Sp = _attn_pad_len(S, causal)
        o, lse = hip_ext().attn_fwd(_pad_seq(q, Sp), _pad_seq(k, Sp),
                                    _pad_seq(v, Sp), scale, causal)
        if Sp != S:
            o = o[:, :, :S].contiguous()
            lse = lse[:, :, :S].contiguous()
        return o, lse
    s = _ref_scores(q, k, scale, causal)
    lse = torch.logsumexp(s, dim=-1)
    p = torch.exp(s - lse.unsqueeze(-1))
    G = q.size(1) // k.size(1)
    vf = v.float().repeat_interleave(G, dim=1)
    return torch.matmul(p, vf).to(q.dtype), lse


def attn_bwd_raw(q, k, v, o, dout, lse, scale, causal=True):
    """Non-autograd flash backward against a GLOBAL (o, lse).

    The softmax statistics may cover MORE kv than (k, v) — exactly the
    per-chunk backward a ring-attention pass needs: p is formed from the
    global lse and delta from the global o, so each chunk's (dq, dk, dv)
    contribution sums to the true gradient. Mirrors the HIP kernels'
    formula (attention_v2.hip: p = exp(s*scale - lse),
    ds = p * (dp - delta) * scale).
    """
    if q.is_cuda:
        S = q.size(2)
        Sp = _attn_pad_len(S, causal)
        if Sp == S:
            return hip_ext().attn_bwd(q.contiguous(), k.contiguous(),
                                      v.contiguous(), o.contiguous(),
                                      dout.contiguous(), lse, scale, causal)
        # padded rows: dout rows are zero so ds == 0 there; padded lse
        # rows must be finite for exp() — zero q/k gives lse=log(n), but
        # here lse came from the CALLER (global ring statistic) so pad
        # with zeros explicitly (p = exp(0-0) is finite, ds still 0).
        lse_p = lse.new_zeros((lse.size(0), lse.size(1), Sp))
        lse_p[:, :, :S] = lse
        dq, dk, dv = hip_ext().attn_bwd(
            _pad_seq(q, Sp), _pad_seq(k, Sp), _pad_seq(v, Sp),
            _pad_seq(o, Sp), _pad_seq(dout, Sp), lse_p, scale, causal)
        return (dq[:, :, :S].contiguous(), dk[:, :, :S].contiguous(),
                dv[:, :, :S].contiguous())
    G = q.size(1) // k.size(1)
    Hkv = k.size(1)
    s = _ref_scores(q, k, scale, causal)
    p = torch.exp(s - lse.unsqueeze(-1))       # global-softmax probs
    kf = k.float().repeat_interleave(G, dim=1)
    vf = v.float().repeat_interleave(G, dim=1)
    dof = dout.float()
    delta = (dof * o.float()).sum(-1)          # [B,H,S]
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    ds = p * (dp - delta.unsqueeze(-1)) * scale
    dq = torch.matmul(ds, kf).to(q.dtype)
    B, H, S, D = q.shape
    Skv = k.size(2)
    dsg = ds.view(B, Hkv, G, S, Skv)
    qg = q.float().view(B, Hkv, G, S, D)
    dog = dof.view(B, Hkv, G, S, D)
    pg = p.view(B, Hkv, G, S, Skv)
    dk = torch.einsum("bhgsk,bhgsd->bhkd", dsg, qg).to(k.dtype)
    dv = torch.einsum("bhgsk,bhgsd->bhkd", pg, dog).to(v.dtype)
    return dq, dk, dv


def attn_decode(q, k_cache, v_cache, length, scale):
    """Split-K flash-decode: one query row per (b, h) against the first
    ``length`` positions of the preallocated KV cache (no slicing
    copies — the kernel reads at the cache's full stride).

    q: [B, H, 1, 128] bf16; k_cache/v_cache: [B, Hkv, Lmax, 128] bf16.
    GPU -> decode.hip (grid (splits, H, B), memory-bound streaming);
    CPU -> fp32 reference.
    """
    if q.is_cuda:
        return hip_ext().attn_decode(q.contiguous(), k_cache, v_cache,
                                     length, scale)
    return attention_ref(q, k_cache[:, :, :length],
                         v_cache[:, :, :length], scale, causal=False)


def attn_decode_varlen(q, k_cache, v_cache, lengths, scale,
                       max_len=None):
    """Flash-decode with a per-sequence valid prefix (continuous
    batching: slots decode at different positions; length 0 marks an
    inactive slot, whose output row is zero).

    q: [B, H, 1, 128]; lengths: int32 [B] (GPU) or list/IntTensor.
    ``max_len`` bounds the split count; passing the cache capacity makes
    the call hipGraph-capturable (no host read of lengths — the kernel
    chunks from the device lengths, empty splits cost nothing).
    """
    import torch as _torch

    if q.is_cuda:
        if not _torch.is_tensor(lengths):
            lengths = _torch.tensor(lengths, dtype=_torch.int32,
                                    device=q.device)
        lengths = lengths.to(device=q.device, dtype=_torch.int32)
        if max_len is None:
            max_len = int(lengths.max().item())
        return hip_ext().attn_decode_varlen(
            q.contiguous(), k_cache, v_cache, lengths, max_len, scale)
    outs = []
    for b in range(q.size(0)):
        n = int(lengths[b])
        if n == 0:
            outs.append(_torch.zeros_like(q[b:b + 1]))
            continue
        outs.append(attention_ref(q[b:b + 1], k_cache[b:b + 1, :, :n],
                                  v_cache[b:b + 1, :, :n], scale,
                                  causal=False))
    return _torch.cat(outs, 0)


# ================================ adam =====================================
def adamw_step(p, g, m, v, step, lr, beta1=0.9, beta2=0.95, eps=1e-8,
               weight_decay=0.1, master=None, grad_scale=1.0):
    """Fused AdamW on one tensor (bf16 or fp32 params, fp32 m/v)."""
    if p.is_cuda:
        hip_ext().adamw(p, g, m, v, master, lr, beta1, beta2, eps,
                        weight_decay, step, grad_scale)
        return
    # CPU reference
    gf = g.float() * grad_scale
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    c1 = 1 / (1 - beta1 ** step)
    c2 = 1 / (1 - beta2 ** step)
    src = master if master is not None else p.float()
    upd = src - lr * ((m * c1) / ((v * c2).sqrt() + eps)
                      + weight_decay * src)
    if master is not None:
        master.copy_(upd)
    p.copy_(upd.to(p.dtype))


def decode_tokens_u16(raw):
    """Packed little-endian uint16 token bytes -> int64 token ids.

    raw: uint8 tensor (GPU: one-pass decode kernel, so the host->device
    copy moves 2 B/token instead of 8; CPU: numpy view). The data-pipe
    decode for pretraining shards (SURVEY §2.4 foreach data path).
    """
    import torch as _torch

    if raw.is_cuda:
        return hip_ext().decode_tokens_u16(raw.contiguous())
    return raw.view(_torch.int16).to(_torch.int64) & 0xFFFF


def add_bf16(a, b):
    if a.is_cuda:
        return hip_ext().add_bf16(a.contiguous(), b.contiguous())
    return a + b
