// Flash attention forward v2: 8-wave 32x32-MFMA swapped-QK^T structure
// (guide §B "8-warp 32x32 ladder"). Key ideas vs v1:
//  * mfma_f32_32x32x16_bf16 (higher ceiling than 16x16 on gfx950);
//  * SWAPPED QK^T: compute S^T = K·Q^T so each lane's C-register column is
//    its OWN q row -> online softmax is entirely in-register (serial max
//    over 32 regs + one shfl_xor(32)), no cross-lane group reductions and
//    no LDS round-trip for P;
//  * P -> PV A-fragments assembled in-register via packed bf16 pairs + one
//    shfl_xor(32) exchange with the partner half-wave;
//  * Q lives in registers (8 x bf16x8 per lane = its own q row);
//  * LDS only holds K (row-major, swizzled) and V^T (swizzled): 32 KiB.
//
// Layouts (probed on HW by dbg_mfma32_kernel):
//  A (32x16): lane holds A[lane&31][(lane>>5)*8 + i]
//  B (16x32): lane holds B[(lane>>5)*8 + i][lane&31]
//  C (32x32): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//
// Shapes: q [B,H,S,128], kv [B,Hkv,S,128], causal, S % 256 == 0 (the
// wrapper falls back to v1 otherwise).

#include "common.h"

typedef __attribute__((ext_vector_type(16))) float f16f;

DEV f16f mfma32(bf16x8 a, bf16x8 b, f16f c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

// pack two fp32 into one u32 of two bf16 (lo, hi): single
// v_cvt_pk_bf16_f32 on gfx950 (RNE) — the manual shift/round sequence
// was ~12 VALU ops and the P->A-fragment repack runs twice per subtile
// in every attention kernel
DEV unsigned int pack_bf2(float lo, float hi) {
  union { __hip_bfloat162 h; unsigned int u; } c;
  c.h = __float22bfloat162_rn(make_float2(lo, hi));
  return c.u;
}

// Turn a per-lane C-tile column (st[2][16]: lane holds values X[row][mycol]
// for rows (r&3)+8*(r>>2)+4*hi of two 32-row sub-tiles) into MFMA
// A-fragments with m = mycol and k = the row dimension: pack bf16 pairs,
// exchange with the partner half-wave (shfl_xor 32), assemble 4 k-steps.
DEV void col_to_afrags(const f16f st[2], bf16x8 pa[4], int hi) {
  unsigned int pk[16], xp[16];
#pragma unroll
  for (int t = 0; t < 2; ++t)
#pragma unroll
    for (int j2 = 0; j2 < 8; ++j2)
      pk[t * 8 + j2] = pack_bf2(st[t][j2 * 2], st[t][j2 * 2 + 1]);
#pragma unroll
  for (int i = 0; i < 16; ++i) xp[i] = __shfl_xor(pk[i], 32, WAVE);
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    union {
      unsigned int w[4];
      bf16x8 v;
    } u;
    if (hi == 0) {
      u.w[0] = pk[4 * ks];
      u.w[1] = pk[4 * ks + 1];
      u.w[2] = xp[4 * ks];
      u.w[3] = xp[4 * ks + 1];
    } else {
      u.w[0] = xp[4 * ks + 2];
      u.w[1] = xp[4 * ks + 3];
      u.w[2] = pk[4 * ks + 2];
      u.w[3] = pk[4 * ks + 3];
    }
    pa[ks] = u.v;
  }
}


// single-subtile variant of col_to_afrags: one 32-row C sub-tile -> 2
// A-fragment k-steps (k = 0..31 of the row dimension)
DEV void col_to_afrags1(const f16f& st, bf16x8 pa[2], int hi) {
  unsigned int pk[8], xp[8];
#pragma unroll
  for (int j2 = 0; j2 < 8; ++j2)
    pk[j2] = pack_bf2(st[j2 * 2], st[j2 * 2 + 1]);
#pragma unroll
  for (int i = 0; i < 8; ++i) xp[i] = __shfl_xor(pk[i], 32, WAVE);
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    union { unsigned int w[4]; bf16x8 v; } u;
    if (hi == 0) {
      u.w[0] = pk[4 * ks];
      u.w[1] = pk[4 * ks + 1];
      u.w[2] = xp[4 * ks];
      u.w[3] = xp[4 * ks + 1];
    } else {
      u.w[0] = xp[4 * ks + 2];
      u.w[1] = xp[4 * ks + 3];
      u.w[2] = pk[4 * ks + 2];
      u.w[3] = pk[4 * ks + 3];
    }
    pa[ks] = u.v;
  }
}

// ================= panel layout + hardware transpose reads =================
// A [ROWS][128] bf16 tile stored as 8 panels of [ROWS][16]: element
// (row, d) lives at (d>>4)*ROWS*16 + row*16 + (d&15). Properties:
//  * staging from row-major global is pure vec8 -> vec8 (no scalar scatter);
//  * A-fragment reads (row fixed per lane, 8 contiguous d) are one
//    16-byte read: panel (d0>>4), offset row*16 + (d0&8);
//  * B-fragments with K = the row dimension come from ds_read_b64_tr_b16:
//    the HW pattern elem = base + (l&15) + j*16 + (l>>4)*64 walks column
//    (l&15) of a [4][16] subtile, which in panel storage is exactly
//    4 consecutive rows at one d-column — solve base per lane so lane
//    l = 32*hi + l32 receives rows q0+hi*8+j at column n*32+l32.

template <int ROWS, int BLOCK>
DEV void stage_panel(short* dst, const short* src,
                     long long src_row_stride) {
  constexpr int TOT = ROWS * 16;  // vec8 slots
#pragma unroll
  for (int it = 0; it < (TOT + BLOCK - 1) / BLOCK; ++it) {
    const int idx = it * BLOCK + threadIdx.x;
    if (idx < TOT) {
      const int r = idx % ROWS, c8 = idx / ROWS;
      bf16x8 v = *(const bf16x8*)(src + r * src_row_stride + c8 * 8);
      *(bf16x8*)(dst + (c8 >> 1) * (ROWS * 16) + r * 16 + (c8 & 1) * 8) =
          v;
    }
  }
}

template <int ROWS>
DEV bf16x8 frag8_panel(const short* tile, int row, int d0) {
  return *(const bf16x8*)(tile + (d0 >> 4) * (ROWS * 16) + row * 16
                          + (d0 & 15));
}

// issue one tr-read: returns 2 VGPRs holding rows (base_q .. base_q+3) at
// this lane's column; caller must s_waitcnt lgkmcnt + sched_barrier before
// consuming (rule #18).
DEV unsigned long long tr_read(const short* lds_ptr) {
  // generic pointers to __shared__ carry the LDS offset in the low 32
  // bits; ds_* instructions take that 32-bit address
  unsigned int addr = (unsigned int)(unsigned long long)lds_ptr;
  unsigned long long out;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(out) : "v"(addr));
  return out;
}

template <int BLOCK>  // BLOCK = 512 (8 waves)
__global__ __launch_bounds__(512) void attn_fwd_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse, int B, int H, int Hkv, int S, float scale,
    int causal) {
  constexpr int BQ = 256, BKV = 64;  // 8 waves x 32 q rows
  // double-buffered K/V tiles in the 16-wide PANEL layout [8][64][16]:
  // staging is pure vec8 writes, K A-fragments are contiguous panel
  // reads, V B-fragments come from ds_read_b64_tr_b16 (no transposed
  // copy). Stage j+1 overlaps compute on j (T14 async-stage split).
  __shared__ short kp[2][BKV * ATT_D];
  __shared__ short vp[2][BKV * ATT_D];

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (H / Hkv);
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l32 = lane & 31;
  const int hi = lane >> 5;

  const int my_qrow = qb * BQ + wid * 32 + l32;  // this lane's q row
  const long long hoff = ((long long)b * H + h) * S;
  const long long qoff = (hoff + qb * BQ) * ATT_D;
  const long long kvoff0 = ((long long)b * Hkv + hk) * S * ATT_D;

  // Q row in registers: q_reg[s] = Q[my_qrow][s*16 + hi*8 .. +8]
  bf16x8 q_reg[8];
  {
    const short* qrow = q + (hoff + my_qrow) * ATT_D;
#pragma unroll
    for (int s = 0; s < 8; ++s)
      q_reg[s] = *(const bf16x8*)(qrow + s * 16 + hi * 8);
  }

  float m_run = -INFINITY;
  float l_run = 0.f;
  f16f acc_o[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) acc_o[n] = (f16f){};

  // per-thread staging slices: 2 x bf16x8 of K and of V per tile
  const int tid = threadIdx.x;
  bf16x8 stg_k[2], stg_v[2];
  int stg_r[2], stg_c8[2];
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = u * BLOCK + tid;
    stg_r[u] = idx / 16;
    stg_c8[u] = idx % 16;
  }

#define MFX_STAGE_LOAD(jj)                                               \
  {                                                                      \
    const short* kb_ = k + kvoff0 + (long long)(jj) * BKV * ATT_D;       \
    const short* vb_ = v + kvoff0 + (long long)(jj) * BKV * ATT_D;       \
    _Pragma("unroll") for (int u = 0; u < 2; ++u) {                      \
      stg_k[u] = *(const bf16x8*)(kb_ + stg_r[u] * ATT_D + stg_c8[u] * 8);\
      stg_v[u] = *(const bf16x8*)(vb_ + stg_r[u] * ATT_D + stg_c8[u] * 8);\
    }                                                                    \
  }

#define MFX_STAGE_WRITE(buf)                                             \
  {                                                                      \
    _Pragma("unroll") for (int u = 0; u < 2; ++u) {                      \
      const int pel_ = (stg_c8[u] >> 1) * (BKV * 16) + stg_r[u] * 16     \
                       + (stg_c8[u] & 1) * 8;                            \
      *(bf16x8*)(kp[buf] + pel_) = stg_k[u];                             \
      *(bf16x8*)(vp[buf] + pel_) = stg_v[u];                             \
    }                                                                    \
  }

  const int kv_tiles =
      causal ? (qb * BQ + BQ) / BKV : S / BKV;  // causal bound incl diag
  MFX_STAGE_LOAD(0);
  MFX_STAGE_WRITE(0);
  __syncthreads();
  for (int j = 0; j < kv_tiles; ++j) {
    const int cur = j & 1;
    if (j + 1 < kv_tiles) MFX_STAGE_LOAD(j + 1);  // issue loads early

    // wave-uniform skip: this wave's rows are all below the tile's kv
    // range (fully masked) — staging + barrier still run below
    const bool active = !causal || (j * BKV <= qb * BQ + wid * 32 + 31);
    if (active) {

    // ---- S^T = K (64x128) @ Q^T: two 32-kv sub-tiles ----
    f16f st[2];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      st[t] = (f16f){};
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        bf16x8 kf = frag8_panel<BKV>(kp[cur], t * 32 + l32,
                                     s * 16 + hi * 8);
        st[t] = mfma32(kf, q_reg[s], st[t]);
      }
    }

    // ---- scale + causal mask + in-register online softmax ----
    // lane's value (t, r) is S[my_qrow][kv = j*64 + t*32 + krow(r,hi)]
    const bool diag = causal && (j * BKV + BKV > qb * BQ);
    float pmax = -INFINITY;
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float sv = st[t][r] * scale;
        if (diag) {
          const int kvg = j * BKV + t * 32 + (r & 3) + 8 * (r >> 2)
                          + 4 * hi;
          if (kvg > my_qrow) sv = -INFINITY;
        }
        st[t][r] = sv;
        pmax = fmaxf(pmax, sv);
      }
    }
    pmax = fmaxf(pmax, __shfl_xor(pmax, 32, WAVE));
    const float newm = fmaxf(m_run, pmax);
    const float rescale = __expf(m_run - newm);
    m_run = newm;

    float psum = 0.f;
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float e = __expf(st[t][r] - newm);
        st[t][r] = e;
        psum += e;
      }
    psum += __shfl_xor(psum, 32, WAVE);
    l_run = l_run * rescale + psum;

    // ---- rescale O: factor for row qr = (r&3)+8*(r>>2)+4*hi+wid*32 is
    // held by lanes with l32 == qr-local; broadcast via shfl ----
    {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int src_row = (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float f = __shfl(rescale, src_row + 32 * hi, WAVE);
#pragma unroll
        for (int n = 0; n < 4; ++n) acc_o[n][r] *= f;
      }
    }

    // ---- P -> bf16 A-fragments via packed pairs + partner exchange ----
    bf16x8 pa[4];
    col_to_afrags(st, pa, hi);

    // ---- PV: O(32q x 128d) += P(32q x 64kv) @ V(64kv x 128d) ----
    // B-fragments via hardware transpose (see tr_read/panel derivation)
    {
      const int tr_lane_off = (((lane >> 2) & 3) * 16) + (lane & 3) * 4;
      const int tr_panel = ((lane >> 4) & 1) * (BKV * 16);
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        union { unsigned long long u[2]; bf16x8 v; } bfr[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int base = n * 2 * (BKV * 16) + tr_panel
                           + (ks * 16 + hi * 8) * 16 + tr_lane_off;
          bfr[n].u[0] = tr_read(vp[cur] + base);
          bfr[n].u[1] = tr_read(vp[cur] + base + 4 * 16);
        }
        asm volatile("s_waitcnt lgkmcnt(0)");
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc_o[n] = mfma32(pa[ks], bfr[n].v, acc_o[n]);
      }
    }
    }  // active

    if (j + 1 < kv_tiles) MFX_STAGE_WRITE(cur ^ 1);
    __syncthreads();
  }
#undef MFX_STAGE_LOAD
#undef MFX_STAGE_WRITE

  // ---- epilogue: O /= l (row-matched), write bf16 + lse ----
  float inv_l = 1.f / l_run;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const float f = __shfl(inv_l, row_local + 32 * hi, WAVE);
    const long long obase =
        qoff + (long long)(wid * 32 + row_local) * ATT_D;
#pragma unroll
    for (int n = 0; n < 4; ++n)
      o[obase + n * 32 + l32] = f2bf(acc_o[n][r] * f);
  }
  if (hi == 0)
    lse[hoff + my_qrow] = m_run + __logf(l_run);
}

// ===================== BACKWARD v2: dQ ====================================
// Grid (S/256, H, B), 8 waves x 32 q rows. Swapped structure: lane's
// C-column is its OWN q row, so lse/delta are per-lane scalars and
// dS^T -> dS A-fragments use the same col_to_afrags dance. LDS: K rm +
// K^T + V rm (48 KiB).

template <int BLOCK>
__global__ __launch_bounds__(512) void attn_bwd_dq_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, int B, int H, int Hkv, int S, float scale,
    int causal) {
  constexpr int BQ = 256, BKV = 64;
  // panel layout serves both A-fragments (contiguous) and the dQ
  // B-fragments via ds_read_b64_tr_b16 — K^T copy eliminated
  __shared__ short kp[BKV * ATT_D];    // K panels
  __shared__ short vp[BKV * ATT_D];    // V panels

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (H / Hkv);
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l32 = lane & 31;
  const int hi = lane >> 5;

  const int my_qrow = qb * BQ + wid * 32 + l32;
  const long long hoff = ((long long)b * H + h) * S;
  const long long qoff = (hoff + qb * BQ) * ATT_D;
  const long long kvoff0 = ((long long)b * Hkv + hk) * S * ATT_D;

  bf16x8 q_reg[8], do_reg[8];
  {
    const short* qrow = q + (hoff + my_qrow) * ATT_D;
    const short* drow = dout + (hoff + my_qrow) * ATT_D;
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      q_reg[s] = *(const bf16x8*)(qrow + s * 16 + hi * 8);
      do_reg[s] = *(const bf16x8*)(drow + s * 16 + hi * 8);
    }
  }
  const float my_lse = lse[hoff + my_qrow];
  const float my_del = delta[hoff + my_qrow];

  f16f acc_dq[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) acc_dq[n] = (f16f){};

  const int kv_tiles = causal ? (qb * BQ + BQ) / BKV : S / BKV;
  for (int j = 0; j < kv_tiles; ++j) {
    __syncthreads();
    stage_panel<BKV, BLOCK>(kp, k + kvoff0 + (long long)j * BKV * ATT_D,
                            ATT_D);
    stage_panel<BKV, BLOCK>(vp, v + kvoff0 + (long long)j * BKV * ATT_D,
                            ATT_D);
    __syncthreads();
    if (causal && j * BKV > qb * BQ + wid * 32 + 31) continue;

    // per 32-kv sub-tile: S^T, dP^T, dS, dQ — keeps live regs low
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      f16f st = (f16f){}, dpt = (f16f){};
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        bf16x8 kf = frag8_panel<BKV>(kp, t * 32 + l32, s * 16 + hi * 8);
        bf16x8 vf = frag8_panel<BKV>(vp, t * 32 + l32, s * 16 + hi * 8);
        st = mfma32(kf, q_reg[s], st);
        dpt = mfma32(vf, do_reg[s], dpt);
      }
      // dS = P * (dP - delta) * scale, P = exp(S*scale - lse), causal
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvg = j * BKV + t * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float p = 0.f;
        if (!causal || kvg <= my_qrow)
          p = __expf(st[r] * scale - my_lse);
        st[r] = p * (dpt[r] - my_del) * scale;
      }
      // dQ(32q x 128d) += dS(32q x 32kv) @ K(32kv x 128d)
      bf16x8 pa[2];
      col_to_afrags1(st, pa, hi);
      const int tr_lane_off = (((lane >> 2) & 3) * 16) + (lane & 3) * 4;
      const int tr_panel = ((lane >> 4) & 1) * (BKV * 16);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int kv0 = t * 32 + ks * 16;
        union { unsigned long long u[2]; bf16x8 v; } bfr[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int base = n * 2 * (BKV * 16) + tr_panel
                           + (kv0 + hi * 8) * 16 + tr_lane_off;
          bfr[n].u[0] = tr_read(kp + base);
          bfr[n].u[1] = tr_read(kp + base + 4 * 16);
        }
        asm volatile("s_waitcnt lgkmcnt(0)");
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc_dq[n] = mfma32(pa[ks], bfr[n].v, acc_dq[n]);
      }
    }
  }

  // epilogue: C row = q local, col = d
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const long long obase =
        qoff + (long long)(wid * 32 + row_local) * ATT_D;
#pragma unroll
    for (int n = 0; n < 4; ++n)
      dq[obase + n * 32 + l32] = f2bf(acc_dq[n][r]);
  }
}

// ===================== BACKWARD v2: dK/dV =================================
// Grid (S/256, H, B) — one Q-HEAD per block, mirroring the dq kernel's
// structure (which measures 2.2x more TF/s than the round-1 dkdv that
// looped all G q-heads inside a (S/256, Hkv, B) grid: 4x fewer blocks
// with a skewed causal trapezoid starved/imbalanced the chip). Each
// block accumulates its head's (dK, dV) contribution in registers and
// writes FP32 PARTIALS laid out [B, H, S, D]; dkdv_reduce_kernel sums
// the G partials per kv head and converts to bf16 — numerically
// identical to the old in-register fp32 accumulation across g. 8 waves
// x 32 kv rows; K/V rows stay register-resident (the compiler hoists
// them at the 256-VGPR budget, 0 spill — probed locally with
// -Rpass-analysis=kernel-resource-usage). LDS: Q + dO panels (32 KiB).

template <int BLOCK>
__global__ __launch_bounds__(512) void attn_bwd_dkdv_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    float* __restrict__ dk_part, float* __restrict__ dv_part,
    int B, int H, int Hkv, int S, float scale, int causal) {
  constexpr int BKVB = 256, BQ2 = 64;
  __shared__ short qp[BQ2 * ATT_D];    // Q panels
  __shared__ short dop[BQ2 * ATT_D];   // dO panels
  __shared__ float lse_s[BQ2];
  __shared__ float del_s[BQ2];

  const int kvb = blockIdx.x;
  const int h = blockIdx.y;           // q head
  const int b = blockIdx.z;
  const int G = H / Hkv;
  const int hk = h / G;               // kv head this block feeds
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l32 = lane & 31;
  const int hi = lane >> 5;

  const int my_kvrow = kvb * BKVB + wid * 32 + l32;
  const long long kvoff =
      (((long long)b * Hkv + hk) * S + kvb * BKVB) * ATT_D;

  // K/V rows for this wave's 32 kv rows; register-resident across the
  // whole q loop at the 256-VGPR budget
  const short* krow = k + kvoff + (long long)(wid * 32 + l32) * ATT_D;
  const short* vrow = v + kvoff + (long long)(wid * 32 + l32) * ATT_D;

  f16f acc_dk[4], acc_dv[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    acc_dk[n] = (f16f){};
    acc_dv[n] = (f16f){};
  }

  const long long hoff = ((long long)b * H + h) * S;
  const int jq0 = causal ? (kvb * BKVB) / BQ2 : 0;
  const int nq = S / BQ2;
  for (int jq = jq0; jq < nq; ++jq) {
    __syncthreads();
    const short* qsrc = q + (hoff + (long long)jq * BQ2) * ATT_D;
    const short* dsrc = dout + (hoff + (long long)jq * BQ2) * ATT_D;
    stage_panel<BQ2, BLOCK>(qp, qsrc, ATT_D);
    stage_panel<BQ2, BLOCK>(dop, dsrc, ATT_D);
    if (threadIdx.x < BQ2) {
      lse_s[threadIdx.x] = lse[hoff + jq * BQ2 + threadIdx.x];
      del_s[threadIdx.x] = delta[hoff + jq * BQ2 + threadIdx.x];
    }
    __syncthreads();
    // wave-uniform skip: all of this wave's kv rows above every q row
    if (causal && jq * BQ2 + BQ2 - 1 < kvb * BKVB + wid * 32) continue;

    // per 32-q sub-tile: S, dP, P/dS, dV, dK — keeps live regs low
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      f16f st = (f16f){}, dpt = (f16f){};
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        bf16x8 qf = frag8_panel<BQ2>(qp, t * 32 + l32, s * 16 + hi * 8);
        bf16x8 df = frag8_panel<BQ2>(dop, t * 32 + l32, s * 16 + hi * 8);
        bf16x8 kr = *(const bf16x8*)(krow + s * 16 + hi * 8);
        bf16x8 vr = *(const bf16x8*)(vrow + s * 16 + hi * 8);
        st = mfma32(qf, kr, st);
        dpt = mfma32(df, vr, dpt);
      }
      // P (into st) with causal mask q >= kv; dS (into dpt). lse/delta
      // come as f32x4 GROUP loads: the per-element lse_s[qrl] reads were
      // 64 scalar ds_read_b32 per tile (the dq kernel holds its lse in
      // a register) — the disassembly showed them, plus their waitcnts,
      // matching the mfma count, which is where dkdv's efficiency gap
      // vs dq was hiding. qrl = t*32 + (r&3) + 8*(r>>2) + 4*hi, so
      // r = g*4+j walks 4 consecutive floats at t*32 + 8g + 4hi.
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const int q0l = t * 32 + g * 8 + 4 * hi;
        const f32x4 lv = *(const f32x4*)&lse_s[q0l];
        const f32x4 dl = *(const f32x4*)&del_s[q0l];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int r = g * 4 + j;
          const int qrg = jq * BQ2 + q0l + j;
          float p = 0.f;
          if (!causal || qrg >= my_kvrow)
            p = __expf(st[r] * scale - lv[j]);
          st[r] = p;
          dpt[r] = p * (dpt[r] - dl[j]) * scale;
        }
      }
      // ds_read_b64_tr_b16 semantics: each lane loads 64b at its OWN
      // address; the HW transposes 16-bit elements within each 16-lane
      // group (a [4][16] tile). To make lane l=32*hi+l32 RECEIVE rows
      // q0+hi*8+j at panel column n*32+l32, lane l must LOAD the 4
      // elements of row q0+hi*8+((l>>2)&3) at col offset 4*(l&3):
      const int tr_lane_off = (((lane >> 2) & 3) * 16) + (lane & 3) * 4;
      const int tr_panel = ((lane >> 4) & 1) * (BQ2 * 16);
      // dV(32kv x 128d) += P^T(32kv x 32q) @ dO(32q x 128d), then
      // dK += dS^T @ Q — FOUR ds_read_b64_tr groups, software-pipelined
      // with COUNTED waits: group g+1's 8 DS reads issue before waiting
      // on group g (DS returns in order, so lgkmcnt(8) = "the earlier 8
      // are done"), so the transpose-read latency of every group but
      // the last hides under the previous group's MFMAs. The PMC run
      // (gpurun_out/pmc) showed both backward kernels stalled on these
      // full-stop groups, not on LDS bandwidth (LdsUtil 12%, conflicts
      // 4%).
      bf16x8 pa[2], pak[2];
      col_to_afrags1(st, pa, hi);
      union { unsigned long long u[2]; bf16x8 v; } bfr[2][4];
      const short* g_tile[4] = {dop, dop, qp, qp};
      int g_q0[4];
      g_q0[0] = t * 32;
      g_q0[1] = t * 32 + 16;
      g_q0[2] = t * 32;
      g_q0[3] = t * 32 + 16;
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int base = n * 2 * (BQ2 * 16) + tr_panel
                         + (g_q0[0] + hi * 8) * 16 + tr_lane_off;
        bfr[0][n].u[0] = tr_read(g_tile[0] + base);
        bfr[0][n].u[1] = tr_read(g_tile[0] + base + 4 * 16);
      }
      col_to_afrags1(dpt, pak, hi);     // VALU under the first DS group
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const int cur = g & 1;
        if (g < 3) {
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int base = n * 2 * (BQ2 * 16) + tr_panel
                             + (g_q0[g + 1] + hi * 8) * 16 + tr_lane_off;
            bfr[cur ^ 1][n].u[0] = tr_read(g_tile[g + 1] + base);
            bfr[cur ^ 1][n].u[1] = tr_read(g_tile[g + 1] + base + 4 * 16);
          }
          asm volatile("s_waitcnt lgkmcnt(8)");
        } else {
          asm volatile("s_waitcnt lgkmcnt(0)");
        }
        __builtin_amdgcn_sched_barrier(0);
        const bf16x8 a = (g < 2) ? pa[g] : pak[g - 2];
        f16f* acc = (g < 2) ? acc_dv : acc_dk;
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[n] = mfma32(a, bfr[cur][n].v, acc[n]);
      }
    }
  }

  // epilogue: fp32 partials at [b, h, kv row, d]
  const long long poff = (hoff + (long long)kvb * BKVB) * ATT_D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const long long obase =
        poff + (long long)(wid * 32 + row_local) * ATT_D;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      dk_part[obase + n * 32 + l32] = acc_dk[n][r];
      dv_part[obase + n * 32 + l32] = acc_dv[n][r];
    }
  }
}

// Sum the G per-q-head fp32 partials into the kv head's bf16 (dK, dV).
// Memory-bound; vectorized f32x4 loads, grid-stride.
__global__ void dkdv_reduce_kernel(const float* __restrict__ dk_part,
                                   const float* __restrict__ dv_part,
                                   short* __restrict__ dk,
                                   short* __restrict__ dv,
                                   long long n4_kv, int G,
                                   long long head_elems) {
  // n4_kv = B*Hkv*S*D/4. Flat kv vec4-index i4 -> kv slot (b*Hkv + hk);
  // its G partials live at heads slot*G + g of the [B, H, S, D] buffer
  // (h = hk*G + g).
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i4 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i4 < n4_kv; i4 += stride) {
    const long long i = i4 * 4;
    const long long slot = i / head_elems;
    const long long rem = i - slot * head_elems;
    const float* kp = dk_part + slot * G * head_elems + rem;
    const float* vp = dv_part + slot * G * head_elems + rem;
    f32x4 ks = *(const f32x4*)kp;
    f32x4 vs = *(const f32x4*)vp;
    for (int g = 1; g < G; ++g) {
      const f32x4 k2 = *(const f32x4*)(kp + (long long)g * head_elems);
      const f32x4 v2 = *(const f32x4*)(vp + (long long)g * head_elems);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        ks[j] += k2[j];
        vs[j] += v2[j];
      }
    }
    bf16x4 ko, vo;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      ko[j] = f2bf(ks[j]);
      vo[j] = f2bf(vs[j]);
    }
    *(bf16x4*)(dk + i) = ko;
    *(bf16x4*)(dv + i) = vo;
  }
}

// ============== BACKWARD v2 (split variant): dV-only / dK-only ============
// Experiment (MFX_ATTN_DKDV_SPLIT=1): the fused dkdv kernel holds 128
// live accumulator VGPRs (dk+dv) vs the dq kernel's 64, and measures
// 187-260 TF/s vs dq's 357 at identical occupancy with every
// latency-hiding experiment flat — pointing at scheduler ILP starved
// by register pressure. Splitting halves the live accumulators per
// kernel at the cost of recomputing S^T (5 GEMM-equivalents total vs
// 4): dV needs only S^T -> P -> P^T·dO (K resident, no V, no delta);
// dK needs S^T, dP^T, dS -> dS^T·Q (K+V resident). waves_per_eu(3) on
// the dV kernel caps it at 170 VGPRs for 3 waves/SIMD.

template <int BLOCK>
__global__ __launch_bounds__(512)
__attribute__((amdgpu_waves_per_eu(3))) void attn_bwd_dv_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ dout, const float* __restrict__ lse,
    float* __restrict__ dv_part,
    int B, int H, int Hkv, int S, float scale, int causal) {
  constexpr int BKVB = 256, BQ2 = 64;
  __shared__ short qp[BQ2 * ATT_D];
  __shared__ short dop[BQ2 * ATT_D];
  __shared__ float lse_s[BQ2];

  const int kvb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int G = H / Hkv;
  const int hk = h / G;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l32 = lane & 31;
  const int hi = lane >> 5;

  const int my_kvrow = kvb * BKVB + wid * 32 + l32;
  const long long kvoff =
      (((long long)b * Hkv + hk) * S + kvb * BKVB) * ATT_D;
  const short* krow = k + kvoff + (long long)(wid * 32 + l32) * ATT_D;

  f16f acc_dv[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) acc_dv[n] = (f16f){};

  const long long hoff = ((long long)b * H + h) * S;
  const int jq0 = causal ? (kvb * BKVB) / BQ2 : 0;
  const int nq = S / BQ2;
  for (int jq = jq0; jq < nq; ++jq) {
    __syncthreads();
    const short* qsrc = q + (hoff + (long long)jq * BQ2) * ATT_D;
    const short* dsrc = dout + (hoff + (long long)jq * BQ2) * ATT_D;
    stage_panel<BQ2, BLOCK>(qp, qsrc, ATT_D);
    stage_panel<BQ2, BLOCK>(dop, dsrc, ATT_D);
    if (threadIdx.x < BQ2)
      lse_s[threadIdx.x] = lse[hoff + jq * BQ2 + threadIdx.x];
    __syncthreads();
    if (causal && jq * BQ2 + BQ2 - 1 < kvb * BKVB + wid * 32) continue;

#pragma unroll
    for (int t = 0; t < 2; ++t) {
      f16f st = (f16f){};
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        bf16x8 qf = frag8_panel<BQ2>(qp, t * 32 + l32, s * 16 + hi * 8);
        bf16x8 kr = *(const bf16x8*)(krow + s * 16 + hi * 8);
        st = mfma32(qf, kr, st);
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrl = t * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const int qrg = jq * BQ2 + qrl;
        float p = 0.f;
        if (!causal || qrg >= my_kvrow)
          p = __expf(st[r] * scale - lse_s[qrl]);
        st[r] = p;
      }
      const int tr_lane_off = (((lane >> 2) & 3) * 16) + (lane & 3) * 4;
      const int tr_panel = ((lane >> 4) & 1) * (BQ2 * 16);
      bf16x8 pa[2];
      col_to_afrags1(st, pa, hi);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int q0 = t * 32 + ks * 16;
        union { unsigned long long u[2]; bf16x8 v; } bfr[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int base = n * 2 * (BQ2 * 16) + tr_panel
                           + (q0 + hi * 8) * 16 + tr_lane_off;
          bfr[n].u[0] = tr_read(dop + base);
          bfr[n].u[1] = tr_read(dop + base + 4 * 16);
        }
        asm volatile("s_waitcnt lgkmcnt(0)");
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc_dv[n] = mfma32(pa[ks], bfr[n].v, acc_dv[n]);
      }
    }
  }

  const long long poff = (hoff + (long long)kvb * BKVB) * ATT_D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const long long obase =
        poff + (long long)(wid * 32 + row_local) * ATT_D;
#pragma unroll
    for (int n = 0; n < 4; ++n)
      dv_part[obase + n * 32 + l32] = acc_dv[n][r];
  }
}

template <int BLOCK>
__global__ __launch_bounds__(512) void attn_bwd_dk_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    float* __restrict__ dk_part,
    int B, int H, int Hkv, int S, float scale, int causal) {
  constexpr int BKVB = 256, BQ2 = 64;
  __shared__ short qp[BQ2 * ATT_D];
  __shared__ short dop[BQ2 * ATT_D];
  __shared__ float lse_s[BQ2];
  __shared__ float del_s[BQ2];

  const int kvb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int G = H / Hkv;
  const int hk = h / G;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l32 = lane & 31;
  const int hi = lane >> 5;

  const int my_kvrow = kvb * BKVB + wid * 32 + l32;
  const long long kvoff =
      (((long long)b * Hkv + hk) * S + kvb * BKVB) * ATT_D;
  const short* krow = k + kvoff + (long long)(wid * 32 + l32) * ATT_D;
  const short* vrow = v + kvoff + (long long)(wid * 32 + l32) * ATT_D;

  f16f acc_dk[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) acc_dk[n] = (f16f){};

  const long long hoff = ((long long)b * H + h) * S;
  const int jq0 = causal ? (kvb * BKVB) / BQ2 : 0;
  const int nq = S / BQ2;
  for (int jq = jq0; jq < nq; ++jq) {
    __syncthreads();
    const short* qsrc = q + (hoff + (long long)jq * BQ2) * ATT_D;
    const short* dsrc = dout + (hoff + (long long)jq * BQ2) * ATT_D;
    stage_panel<BQ2, BLOCK>(qp, qsrc, ATT_D);
    stage_panel<BQ2, BLOCK>(dop, dsrc, ATT_D);
    if (threadIdx.x < BQ2) {
      lse_s[threadIdx.x] = lse[hoff + jq * BQ2 + threadIdx.x];
      del_s[threadIdx.x] = delta[hoff + jq * BQ2 + threadIdx.x];
    }
    __syncthreads();
    if (causal && jq * BQ2 + BQ2 - 1 < kvb * BKVB + wid * 32) continue;

#pragma unroll
    for (int t = 0; t < 2; ++t) {
      f16f st = (f16f){}, dpt = (f16f){};
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        bf16x8 qf = frag8_panel<BQ2>(qp, t * 32 + l32, s * 16 + hi * 8);
        bf16x8 df = frag8_panel<BQ2>(dop, t * 32 + l32, s * 16 + hi * 8);
        bf16x8 kr = *(const bf16x8*)(krow + s * 16 + hi * 8);
        bf16x8 vr = *(const bf16x8*)(vrow + s * 16 + hi * 8);
        st = mfma32(qf, kr, st);
        dpt = mfma32(df, vr, dpt);
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrl = t * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const int qrg = jq * BQ2 + qrl;
        float p = 0.f;
        if (!causal || qrg >= my_kvrow)
          p = __expf(st[r] * scale - lse_s[qrl]);
        dpt[r] = p * (dpt[r] - del_s[qrl]) * scale;
      }
      const int tr_lane_off = (((lane >> 2) & 3) * 16) + (lane & 3) * 4;
      const int tr_panel = ((lane >> 4) & 1) * (BQ2 * 16);
      bf16x8 pa[2];
      col_to_afrags1(dpt, pa, hi);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int q0 = t * 32 + ks * 16;
        union { unsigned long long u[2]; bf16x8 v; } bfr[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int base = n * 2 * (BQ2 * 16) + tr_panel
                           + (q0 + hi * 8) * 16 + tr_lane_off;
          bfr[n].u[0] = tr_read(qp + base);
          bfr[n].u[1] = tr_read(qp + base + 4 * 16);
        }
        asm volatile("s_waitcnt lgkmcnt(0)");
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc_dk[n] = mfma32(pa[ks], bfr[n].v, acc_dk[n]);
      }
    }
  }

  const long long poff = (hoff + (long long)kvb * BKVB) * ATT_D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const long long obase =
        poff + (long long)(wid * 32 + row_local) * ATT_D;
#pragma unroll
    for (int n = 0; n < 4; ++n)
      dk_part[obase + n * 32 + l32] = acc_dk[n][r];
  }
}
