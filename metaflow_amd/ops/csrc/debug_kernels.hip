// Isolation kernels for the attention-backward GEMM microstructures.
// Each replicates one exact code path from attention.hip on arbitrary
// inputs so host-side matmul can verify it.

#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f4_;

// path A: S^T = K(64x128) @ Q^T(128x32) via A-preload + B row frags
__global__ __launch_bounds__(256) void dbg_st_kernel(
    const short* __restrict__ kmat, const short* __restrict__ qmat,
    float* __restrict__ out /* [64][32] */) {
  constexpr int BLOCK = 256;
  __shared__ short kt[64 * 128];
  __shared__ short qtile[32 * 128];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & 63;
  const int lgrp = lane >> 4;
  const int l16 = lane & 15;

  stage_rm<64, BLOCK>(kt, kmat, 128);
  stage_rm<32, BLOCK>(qtile, qmat, 128);
  __syncthreads();

  bf16x8 kf[4];
#pragma unroll
  for (int kk = 0; kk < 4; ++kk)
    kf[kk] = frag8(kt, wid * 16 + l16, kk * 32 + lgrp * 8, 256);

  f4_ st[2];
#pragma unroll
  for (int n = 0; n < 2; ++n) {
    st[n] = (f4_){0, 0, 0, 0};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      bf16x8 qb8 = frag8(qtile, n * 16 + l16, kk * 32 + lgrp * 8, 256);
      st[n] = mfma16(kf[kk], qb8, st[n]);
    }
  }
#pragma unroll
  for (int n = 0; n < 2; ++n)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      out[(wid * 16 + lgrp * 4 + r) * 32 + n * 16 + l16] = st[n][r];
}

// path B: C = P(64x32, given in C-layout regs via ptile roundtrip) @
//         B(32x128, staged transposed) — the exact dV structure
__global__ __launch_bounds__(256) void dbg_dv_kernel(
    const short* __restrict__ pmat /* [64][32] */,
    const short* __restrict__ bmat /* [32][128] */,
    float* __restrict__ out /* [64][128] */) {
  constexpr int BLOCK = 256;
  __shared__ short ptile[64 * 32];
  __shared__ short dott[128 * 32];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & 63;
  const int lgrp = lane >> 4;
  const int l16 = lane & 15;

  stage_tr<32, BLOCK>(dott, bmat, 128);
  __syncthreads();

  // load P into C-layout regs exactly like st[] holds P^T, then roundtrip
  float st[2][4];
#pragma unroll
  for (int n = 0; n < 2; ++n)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      st[n][r] = bf2f(pmat[(wid * 16 + lgrp * 4 + r) * 32 + n * 16 + l16]);

#pragma unroll
  for (int n = 0; n < 2; ++n)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      *(short*)((char*)ptile + swz(wid * 16 + lgrp * 4 + r, n * 16 + l16,
                                   64)) = f2bf(st[n][r]);
  __syncthreads();

  f4_ acc[8];
#pragma unroll
  for (int n = 0; n < 8; ++n) acc[n] = (f4_){0, 0, 0, 0};
  bf16x8 pa = frag8(ptile, wid * 16 + l16, lgrp * 8, 64);
#pragma unroll
  for (int n = 0; n < 8; ++n) {
    bf16x8 db8 = frag8(dott, n * 16 + l16, lgrp * 8, 64);
    acc[n] = mfma16(pa, db8, acc[n]);
  }
#pragma unroll
  for (int n = 0; n < 8; ++n)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      out[(wid * 16 + lgrp * 4 + r) * 128 + n * 16 + l16] = acc[n][r];
}

// probe: mfma_f32_32x32x16_bf16 layout — C = A(32x16) @ B(16x32) with
// assumed layouts: A[lane&31][(lane>>5)*8+i], B[(lane>>5)*8+i][lane&31],
// C: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
typedef __attribute__((ext_vector_type(16))) float f16x_;
__global__ void dbg_mfma32_kernel(const short* __restrict__ a,
                                  const short* __restrict__ b,
                                  float* __restrict__ c /* [32][32] */) {
  const int lane = threadIdx.x & 63;
  const int l32 = lane & 31;
  const int hi = lane >> 5;
  bf16x8 av, bv;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    av[i] = a[l32 * 16 + hi * 8 + i];       // A row-major [32][16]
    bv[i] = b[(hi * 8 + i) * 32 + l32];     // B row-major [16][32]
  }
  f16x_ acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av, bv, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    c[row * 32 + l32] = acc[r];
  }
}
