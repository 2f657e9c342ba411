// Memory-bound training kernels for gfx950: RMSNorm fwd/bwd, RoPE fwd/bwd,
// SwiGLU fwd/bwd. All bf16 I/O with fp32 math, bf16x8 vectorized loads
// (guide G13: scalar bf16 loads cost ~2-2.5x).
//
// No reference counterpart: Netflix/metaflow has zero kernels;
// these implement the @parallel train-step hot path the reference delegates
// to user code.

#include "common.h"

// ---------------------------------------------------------------- RMSNorm
// y[n][h] = x[n][h] * w[h] * rsqrt(mean_h(x^2) + eps); saves inv_rms[n].
// One block (256 threads) per row; rows = B*S, H up to 16384 with bf16x8.

// optional fused residual add: when res != null, operates on s = x + res
// and writes s to sum_out (the new residual stream) — one launch instead
// of add + norm, and s is read back from LDS-fresh L2 instead of recompute.
template <int BLOCK>
__global__ void rmsnorm_fwd_kernel(const short* __restrict__ x,
                                   const short* __restrict__ res,
                                   short* __restrict__ sum_out,
                                   const short* __restrict__ w,
                                   short* __restrict__ y,
                                   float* __restrict__ inv_rms,
                                   int H, float eps) {
  __shared__ float scratch[BLOCK / WAVE];
  const long long row = blockIdx.x;
  const short* xr = x + row * (long long)H;
  short* yr = y + row * (long long)H;
  const bf16x8* rv = res ? (const bf16x8*)(res + row * (long long)H)
                         : nullptr;
  bf16x8* sv = sum_out ? (bf16x8*)(sum_out + row * (long long)H) : nullptr;

  float ssq = 0.f;
  const int HV = H / 8;
  const bf16x8* xv = (const bf16x8*)xr;
  for (int i = threadIdx.x; i < HV; i += BLOCK) {
    bf16x8 v = xv[i];
    if (rv) {
      bf16x8 r = rv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = f2bf(bf2f(v[j]) + bf2f(r[j]));
      sv[i] = v;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]);
      ssq += f * f;
    }
  }
  ssq = block_sum<BLOCK>(ssq, scratch);
  const float inv = rsqrtf(ssq / (float)H + eps);
  if (threadIdx.x == 0 && inv_rms) inv_rms[row] = inv;

  const bf16x8* wv = (const bf16x8*)w;
  const bf16x8* src = rv ? (const bf16x8*)sv : xv;
  bf16x8* yv = (bf16x8*)yr;
  for (int i = threadIdx.x; i < HV; i += BLOCK) {
    bf16x8 v = src[i];
    bf16x8 wk = wv[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bf(bf2f(v[j]) * inv * bf2f(wk[j]));
    yv[i] = o;
  }
}

// dx[n] = w*dy*inv - x*inv^3/H * sum_h(dy*w*x); dw += sum_n(dy*x*inv).
// dw accumulated per-block in LDS (fp32 H <= 8192 -> 32 KiB), one
// atomicAdd per element at the end (guide G12).
// extra_dsum: optional gradient flowing directly into the (fused) sum
// output — added to dx so add+norm backward is also one launch.
template <int BLOCK>
__global__ void rmsnorm_bwd_kernel(const short* __restrict__ x,
                                   const short* __restrict__ w,
                                   const short* __restrict__ dy,
                                   const short* __restrict__ extra_dsum,
                                   const float* __restrict__ inv_rms,
                                   short* __restrict__ dx,
                                   float* __restrict__ dw,  // fp32 accum
                                   int rows, int H) {
  extern __shared__ float lds[];  // H floats for dw partial + BLOCK/64
  float* dw_part = lds;
  float* scratch = lds + H;
  for (int i = threadIdx.x; i < H; i += BLOCK) dw_part[i] = 0.f;
  __syncthreads();

  const int HV = H / 8;
  const bf16x8* wv = (const bf16x8*)w;

  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16x8* xv = (const bf16x8*)(x + row * (long long)H);
    const bf16x8* dyv = (const bf16x8*)(dy + row * (long long)H);
    bf16x8* dxv = (bf16x8*)(dx + row * (long long)H);
    const float inv = inv_rms[row];

    float dot = 0.f;
    for (int i = threadIdx.x; i < HV; i += BLOCK) {
      bf16x8 xi = xv[i], di = dyv[i], wi = wv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += bf2f(di[j]) * bf2f(wi[j]) * bf2f(xi[j]);
    }
    dot = block_sum<BLOCK>(dot, scratch);
    const float c = dot * inv * inv * inv / (float)H;

    const bf16x8* ev = extra_dsum
        ? (const bf16x8*)(extra_dsum + row * (long long)H) : nullptr;
    for (int i = threadIdx.x; i < HV; i += BLOCK) {
      bf16x8 xi = xv[i], di = dyv[i], wi = wv[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf2f(xi[j]), df = bf2f(di[j]);
        float g = df * bf2f(wi[j]) * inv - xf * c;
        if (ev) g += bf2f(ev[i][j]);
        o[j] = f2bf(g);
        dw_part[i * 8 + j] += df * xf * inv;
      }
      dxv[i] = o;
    }
    __syncthreads();  // dw_part reused across rows
  }
  for (int i = threadIdx.x; i < H; i += BLOCK)
    atomicAdd(&dw[i], dw_part[i]);
}

// ------------------------------------------------------------------- RoPE
// Half-rotation (NeoX/Llama-HF) convention on [N, D] rows where each row is
// one (token, head) pair: out[..d] = x1*cos - x2*sin ; out[D/2+d] =
// x2*cos + x1*sin with cos/sin[pos][d] host-precomputed fp32 (guide App B:
// on-device trig turns memory-bound into VALU-bound).
// pos_of_row: row -> position index (seq pos), stride trick avoids a
// lookup table: pos = (row / heads) % seqlen handled by caller via
// rows_per_pos.
typedef __attribute__((ext_vector_type(4))) float f32x4v;

__global__ void rope_kernel(const short* __restrict__ x,
                            short* __restrict__ y,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            long long rows, int D, int rows_per_pos,
                            int seqlen, int pos0, float sin_sign) {
  // vectorized: each thread rotates 8 (x1,x2) pairs -> 2x bf16x8 loads,
  // 2x f32x4 table loads, 2x bf16x8 stores (guide G13)
  const int half = D / 2;
  const int hv = half / 8;  // bf16x8 chunks per half
  const long long total = rows * (long long)hv;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const long long row = idx / hv;
    const int d8 = (int)(idx % hv);
    const int pos = pos0 + (int)((row / rows_per_pos) % seqlen);
    const long long tbase = (long long)pos * half + d8 * 8;
    const long long base = row * (long long)D + d8 * 8;
    bf16x8 v1 = *(const bf16x8*)(x + base);
    bf16x8 v2 = *(const bf16x8*)(x + base + half);
    f32x4v c0 = *(const f32x4v*)(cos_t + tbase);
    f32x4v c1 = *(const f32x4v*)(cos_t + tbase + 4);
    f32x4v s0 = *(const f32x4v*)(sin_t + tbase);
    f32x4v s1 = *(const f32x4v*)(sin_t + tbase + 4);
    bf16x8 o1, o2;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float c = (j < 4 ? c0[j] : c1[j - 4]);
      const float s = (j < 4 ? s0[j] : s1[j - 4]) * sin_sign;
      const float x1 = bf2f(v1[j]);
      const float x2 = bf2f(v2[j]);
      o1[j] = f2bf(x1 * c - x2 * s);
      o2[j] = f2bf(x2 * c + x1 * s);
    }
    *(bf16x8*)(y + base) = o1;
    *(bf16x8*)(y + base + half) = o2;
  }
}

// --------------------------------------------- fused QKV split + RoPE
// Consumes the fused QKV GEMM output [B, S, (NQ+2*NKV)*128] directly and
// emits rope'd q [B,NQ,S,128], k [B,NKV,S,128] and copied v [B,NKV,S,128]
// in the attention kernel's bhsd layout — replaces three slice+transpose
// contiguous() copies and two rope launches per decoder layer with ONE
// pass over the bytes. Reads and writes are both bf16x8 along D.
__global__ void rope_qkv_fwd_kernel(
    const short* __restrict__ qkv, short* __restrict__ qo,
    short* __restrict__ ko, short* __restrict__ vo,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    int B, int S, int NQ, int NKV) {
  constexpr int D = 128, half = 64, hv = half / 8;
  const int HT = NQ + 2 * NKV;
  const long long total = (long long)B * S * HT * hv;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const long long row = idx / hv;    // (b*S + s)*HT + h
    const int d8 = (int)(idx % hv);
    const int h = (int)(row % HT);
    const long long bs = row / HT;
    const int s = (int)(bs % S);
    const int b = (int)(bs / S);
    const long long src = row * D + d8 * 8;
    short* dst_base;
    if (h < NQ) {
      dst_base = qo + (((long long)b * NQ + h) * S + s) * D;
    } else if (h < NQ + NKV) {
      dst_base = ko + (((long long)b * NKV + (h - NQ)) * S + s) * D;
    } else {
      dst_base = vo + (((long long)b * NKV + (h - NQ - NKV)) * S + s) * D;
    }
    bf16x8 v1 = *(const bf16x8*)(qkv + src);
    bf16x8 v2 = *(const bf16x8*)(qkv + src + half);
    if (h < NQ + NKV) {
      const long long tbase = (long long)s * half + d8 * 8;
      f32x4v c0 = *(const f32x4v*)(cos_t + tbase);
      f32x4v c1 = *(const f32x4v*)(cos_t + tbase + 4);
      f32x4v s0 = *(const f32x4v*)(sin_t + tbase);
      f32x4v s1 = *(const f32x4v*)(sin_t + tbase + 4);
      bf16x8 o1, o2;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float c = (j < 4 ? c0[j] : c1[j - 4]);
        const float sn = (j < 4 ? s0[j] : s1[j - 4]);
        const float x1 = bf2f(v1[j]);
        const float x2 = bf2f(v2[j]);
        o1[j] = f2bf(x1 * c - x2 * sn);
        o2[j] = f2bf(x2 * c + x1 * sn);
      }
      *(bf16x8*)(dst_base + d8 * 8) = o1;
      *(bf16x8*)(dst_base + d8 * 8 + half) = o2;
    } else {
      *(bf16x8*)(dst_base + d8 * 8) = v1;
      *(bf16x8*)(dst_base + d8 * 8 + half) = v2;
    }
  }
}

// Backward: gathers dq/dk/dv [B,H,S,128] back into the fused GEMM-grad
// layout [B, S, (NQ+2*NKV)*128], applying the inverse rotation to the
// q/k parts (rope with -sin).
__global__ void rope_qkv_bwd_kernel(
    const short* __restrict__ dq, const short* __restrict__ dk,
    const short* __restrict__ dv, short* __restrict__ dqkv,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    int B, int S, int NQ, int NKV) {
  constexpr int D = 128, half = 64, hv = half / 8;
  const int HT = NQ + 2 * NKV;
  const long long total = (long long)B * S * HT * hv;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const long long row = idx / hv;
    const int d8 = (int)(idx % hv);
    const int h = (int)(row % HT);
    const long long bs = row / HT;
    const int s = (int)(bs % S);
    const int b = (int)(bs / S);
    const long long dst = row * D + d8 * 8;
    const short* src_base;
    if (h < NQ) {
      src_base = dq + (((long long)b * NQ + h) * S + s) * D;
    } else if (h < NQ + NKV) {
      src_base = dk + (((long long)b * NKV + (h - NQ)) * S + s) * D;
    } else {
      src_base = dv + (((long long)b * NKV + (h - NQ - NKV)) * S + s) * D;
    }
    bf16x8 v1 = *(const bf16x8*)(src_base + d8 * 8);
    bf16x8 v2 = *(const bf16x8*)(src_base + d8 * 8 + half);
    if (h < NQ + NKV) {
      const long long tbase = (long long)s * half + d8 * 8;
      f32x4v c0 = *(const f32x4v*)(cos_t + tbase);
      f32x4v c1 = *(const f32x4v*)(cos_t + tbase + 4);
      f32x4v s0 = *(const f32x4v*)(sin_t + tbase);
      f32x4v s1 = *(const f32x4v*)(sin_t + tbase + 4);
      bf16x8 o1, o2;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float c = (j < 4 ? c0[j] : c1[j - 4]);
        const float sn = -(j < 4 ? s0[j] : s1[j - 4]);
        const float x1 = bf2f(v1[j]);
        const float x2 = bf2f(v2[j]);
        o1[j] = f2bf(x1 * c - x2 * sn);
        o2[j] = f2bf(x2 * c + x1 * sn);
      }
      *(bf16x8*)(dqkv + dst) = o1;
      *(bf16x8*)(dqkv + dst + half) = o2;
    } else {
      *(bf16x8*)(dqkv + dst) = v1;
      *(bf16x8*)(dqkv + dst + half) = v2;
    }
  }
}

// ---------------------------------------------------------------- SwiGLU
// y = silu(g) * u ; dg = dy * u * silu'(g) ; du = dy * silu(g)
__global__ void swiglu_fwd_kernel(const short* __restrict__ g,
                                  const short* __restrict__ u,
                                  short* __restrict__ y, long long n8) {
  const bf16x8* gv = (const bf16x8*)g;
  const bf16x8* uv = (const bf16x8*)u;
  bf16x8* yv = (bf16x8*)y;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += stride) {
    bf16x8 gi = gv[i], ui = uv[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(gi[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      o[j] = f2bf(gf * sig * bf2f(ui[j]));
    }
    yv[i] = o;
  }
}

__global__ void swiglu_bwd_kernel(const short* __restrict__ g,
                                  const short* __restrict__ u,
                                  const short* __restrict__ dy,
                                  short* __restrict__ dg,
                                  short* __restrict__ du, long long n8) {
  const bf16x8* gv = (const bf16x8*)g;
  const bf16x8* uv = (const bf16x8*)u;
  const bf16x8* dyv = (const bf16x8*)dy;
  bf16x8* dgv = (bf16x8*)dg;
  bf16x8* duv = (bf16x8*)du;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += stride) {
    bf16x8 gi = gv[i], ui = uv[i], di = dyv[i], og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(gi[j]), uf = bf2f(ui[j]), df = bf2f(di[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      og[j] = f2bf(df * uf * (sig + silu * (1.f - sig)));
      ou[j] = f2bf(df * silu);
    }
    dgv[i] = og;
    duv[i] = ou;
  }
}

// ------------------------------------------------------- residual add
// fused y = a + b (bf16), used for residual streams to avoid extra eager
// kernels in the step
__global__ void add_bf16_kernel(const short* __restrict__ a,
                                const short* __restrict__ b,
                                short* __restrict__ y, long long n8) {
  const bf16x8* av = (const bf16x8*)a;
  const bf16x8* bv = (const bf16x8*)b;
  bf16x8* yv = (bf16x8*)y;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += stride) {
    bf16x8 x = av[i], z = bv[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(x[j]) + bf2f(z[j]));
    yv[i] = o;
  }
}

// ------------------------------------------------- fused-layout SwiGLU
// gateup: [N, 2I] rows holding [gate | up] halves (the fused MLP GEMM
// output); avoids slice copies entirely — bwd writes d_gateup in the
// same fused layout, feeding the fused GEMM backward directly.
__global__ void swiglu_gu_fwd_kernel(const short* __restrict__ gu,
                                     short* __restrict__ y,
                                     long long rows, int I) {
  const int iv = I / 8;
  const long long total = rows * iv;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const long long row = idx / iv;
    const int c8 = (int)(idx % iv);
    const long long base = row * (long long)(2 * I) + c8 * 8;
    bf16x8 g = *(const bf16x8*)(gu + base);
    bf16x8 u = *(const bf16x8*)(gu + base + I);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      o[j] = f2bf(gf * sig * bf2f(u[j]));
    }
    *(bf16x8*)(y + row * (long long)I + c8 * 8) = o;
  }
}

__global__ void swiglu_gu_bwd_kernel(const short* __restrict__ gu,
                                     const short* __restrict__ dy,
                                     short* __restrict__ dgu,
                                     long long rows, int I) {
  const int iv = I / 8;
  const long long total = rows * iv;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const long long row = idx / iv;
    const int c8 = (int)(idx % iv);
    const long long base = row * (long long)(2 * I) + c8 * 8;
    bf16x8 g = *(const bf16x8*)(gu + base);
    bf16x8 u = *(const bf16x8*)(gu + base + I);
    bf16x8 d = *(const bf16x8*)(dy + row * (long long)I + c8 * 8);
    bf16x8 og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]), uf = bf2f(u[j]), df = bf2f(d[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      og[j] = f2bf(df * uf * (sig + silu * (1.f - sig)));
      ou[j] = f2bf(df * silu);
    }
    *(bf16x8*)(dgu + base) = og;
    *(bf16x8*)(dgu + base + I) = ou;
  }
}

// ------------------------------------------------------ fp8 quantize
// One-pass bf16 -> OCP E4M3 with scale + saturation, via the gfx950
// native packed convert (v_cvt_pk_fp8_f32). The torch composition
// (float() * scale, clamp, .to(fp8)) materializes fp32 intermediates —
// 4+ memory passes that made fp8 GEMMs a net LOSS end-to-end.
__global__ void quant_e4m3_kernel(const short* __restrict__ x,
                                  unsigned char* __restrict__ y,
                                  float scale, long long n8) {
  typedef __attribute__((ext_vector_type(2))) int int2v;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += stride) {
    bf16x8 v = *(const bf16x8*)(x + i * 8);
    float f[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float t = bf2f(v[j]) * scale;
      f[j] = fminf(fmaxf(t, -448.f), 448.f);
    }
    int2v out;
    int w0 = 0, w1 = 0;
    w0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[0], f[1], w0, false);
    w0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[2], f[3], w0, true);
    w1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[4], f[5], w1, false);
    w1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[6], f[7], w1, true);
    out[0] = w0;
    out[1] = w1;
    *(int2v*)(y + i * 8) = out;
  }
}

// E5M2 variant (v_cvt_pk_bf8_f32): wider exponent range for GRADIENT
// tensors in full-fp8 training (dy spans more orders of magnitude than
// activations; e5m2 max normal = 57344). Written at round-2 end for the
// round-3 e5m2-dy experiment — compile-checked, unmeasured.
__global__ void quant_e5m2_kernel(const short* __restrict__ x,
                                  unsigned char* __restrict__ y,
                                  float scale, long long n8) {
  typedef __attribute__((ext_vector_type(2))) int int2v;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += stride) {
    bf16x8 v = *(const bf16x8*)(x + i * 8);
    float f[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float t = bf2f(v[j]) * scale;
      f[j] = fminf(fmaxf(t, -57344.f), 57344.f);
    }
    int2v out;
    int w0 = 0, w1 = 0;
    w0 = __builtin_amdgcn_cvt_pk_bf8_f32(f[0], f[1], w0, false);
    w0 = __builtin_amdgcn_cvt_pk_bf8_f32(f[2], f[3], w0, true);
    w1 = __builtin_amdgcn_cvt_pk_bf8_f32(f[4], f[5], w1, false);
    w1 = __builtin_amdgcn_cvt_pk_bf8_f32(f[6], f[7], w1, true);
    out[0] = w0;
    out[1] = w1;
    *(int2v*)(y + i * 8) = out;
  }
}

// -------------------------------------------------- token decode
// Pretraining shards store tokens as packed uint16 (2 B/token); models
// consume int64. Decoding ON the GPU means the PCIe/H2D copy moves 2
// bytes per token instead of 8 — the foreach data-pipe decode kernel
// (SURVEY §2.4): one memory-bound pass, vectorized 8 tokens/lane.
__global__ void decode_tokens_u16_kernel(
    const unsigned short* __restrict__ x, long long* __restrict__ y,
    long long n8) {
  typedef __attribute__((ext_vector_type(8))) short u16x8;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += stride) {
    u16x8 v = *(const u16x8*)(x + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      y[i * 8 + j] = (long long)(unsigned short)v[j];
  }
}
