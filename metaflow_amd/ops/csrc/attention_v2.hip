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

// pack two fp32 into one u32 of two bf16 (lo, hi)
DEV unsigned int pack_bf2(float lo, float hi) {
  return ((unsigned int)(unsigned short)f2bf(lo))
         | ((unsigned int)(unsigned short)f2bf(hi) << 16);
}

template <int BLOCK>  // BLOCK = 512 (8 waves)
__global__ __launch_bounds__(512) void attn_fwd_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse, int B, int H, int Hkv, int S, float scale) {
  constexpr int BQ = 256, BKV = 64;  // 8 waves x 32 q rows
  __shared__ short kt[BKV * ATT_D];   // K row-major swizzled [64][128]
  __shared__ short vtt[ATT_D * BKV];  // V^T swizzled [128][64]

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

  const int kv_tiles = (qb * BQ + BQ) / BKV;  // causal bound (diag incl.)
  for (int j = 0; j < kv_tiles; ++j) {
    __syncthreads();
    stage_rm<BKV, BLOCK>(kt, k + kvoff0 + (long long)j * BKV * ATT_D,
                         ATT_D);
    stage_tr<BKV, BLOCK>(vtt, v + kvoff0 + (long long)j * BKV * ATT_D,
                         ATT_D);
    __syncthreads();

    // wave-uniform skip: this wave's rows are all below the tile's kv
    // range (fully masked) — barriers above/below still run
    if (j * BKV > qb * BQ + wid * 32 + 31) continue;

    // ---- S^T = K (64x128) @ Q^T: two 32-kv sub-tiles ----
    f16f st[2];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      st[t] = (f16f){};
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        bf16x8 kf = frag8(kt, t * 32 + l32, s * 16 + hi * 8, ATT_D * 2);
        st[t] = mfma32(kf, q_reg[s], st[t]);
      }
    }

    // ---- scale + causal mask + in-register online softmax ----
    // lane's value (t, r) is S[my_qrow][kv = j*64 + t*32 + krow(r,hi)]
    const bool diag = (j * BKV + BKV > qb * BQ);
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
    // own pairs: pk[t*8 + j2] packs kv offsets {t*32 + pairbase(j2) + 4hi}
    unsigned int pk[16], xp[16];
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int j2 = 0; j2 < 8; ++j2) {
        pk[t * 8 + j2] = pack_bf2(st[t][j2 * 2], st[t][j2 * 2 + 1]);
      }
#pragma unroll
    for (int i = 0; i < 16; ++i) xp[i] = __shfl_xor(pk[i], 32, WAVE);

    // A-frag for PV k-step ks (kv = j*64 + ks*16 + hi*8 + i):
    //   hi=0: [pk[4ks], pk[4ks+1], xp[4ks], xp[4ks+1]]   (kv ..0-3,4-7)
    //   hi=1: [xp[4ks+2], xp[4ks+3], pk[4ks+2], pk[4ks+3]] (kv 8-11,12-15)
    bf16x8 pa[4];
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

    // ---- PV: O(32q x 128d) += P(32q x 64kv) @ V(64kv x 128d) ----
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8 vf = frag8(vtt, n * 32 + l32, ks * 16 + hi * 8, BKV * 2);
        acc_o[n] = mfma32(pa[ks], vf, acc_o[n]);
      }
    }
  }

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
