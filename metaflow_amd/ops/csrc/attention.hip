// Shared MFMA-16x16 helpers + attention backward delta preprocess for
// gfx950 (bf16, D=128). The flash fwd/bwd kernels themselves live in
// attention_v2.hip (8-wave 32x32 swapped-QK^T structure); unaligned
// seqlens are zero-padded to the 256 boundary by the Python wrapper
// (exact for causal attention — end-padded keys are causally masked for
// every real query row).
//
// MFMA fragment layout (guide §3, measured m89/m91):
//   C/D: col = lane&15, row = (lane>>4)*4 + reg
//   A (MxK): lane holds A[lane&15][(lane>>4)*8 + i], i = 0..7
//   B (KxN): lane holds B[(lane>>4)*8 + i][lane&15]
// so every A read wants M-major storage [m][k] and every B read wants
// N-major storage [n][k]; tiles are staged in the layout each operand needs.

#include "common.h"

#define ATT_D 128

typedef __attribute__((ext_vector_type(4))) float f4;

DEV f4 mfma16(bf16x8 a, bf16x8 b, f4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// byte offset of element (r, c) in a swizzled LDS tile with ROWB bytes/row
// (rowb must be a power of 2; the XOR is masked so it stays inside the row
// — rowb=64 tiles get a 4-row spread instead of 8, still conflict-free
// enough at 2 lanes/bank)
DEV int swz(int r, int c, int rowb) {
  // spread up to 16 rows across 16-byte slots (256B rows get the full
  // 16-way spread -> 2 lanes/slot for 32-row reads; narrower rows are
  // masked down automatically)
  return r * rowb + (((c) * 2) ^ (((r & 15) << 4) & (rowb - 1)));
}

// load an 8-element fragment from a swizzled tile row (c must be mult of 8)
DEV bf16x8 frag8(const short* tile, int r, int c, int rowb) {
  return *(const bf16x8*)((const char*)tile + swz(r, c, rowb));
}

// 16-lane-group reductions: rows of the C layout live in lanes with equal
// lane>>4; reduce across the 16 lanes of each group.
DEV float group_max(float v) {
  v = fmaxf(v, __shfl_xor(v, 1, WAVE));
  v = fmaxf(v, __shfl_xor(v, 2, WAVE));
  v = fmaxf(v, __shfl_xor(v, 4, WAVE));
  v = fmaxf(v, __shfl_xor(v, 8, WAVE));
  return v;
}

DEV float group_sum(float v) {
  v += __shfl_xor(v, 1, WAVE);
  v += __shfl_xor(v, 2, WAVE);
  v += __shfl_xor(v, 4, WAVE);
  v += __shfl_xor(v, 8, WAVE);
  return v;
}

// stage a [ROWS][128] bf16 tile from global (row-major) into swizzled LDS
template <int ROWS, int BLOCK>
DEV void stage_rm(short* dst, const short* src, long long src_row_stride) {
  constexpr int VPR = ATT_D / 8;  // vec8 per row
  constexpr int TOT = ROWS * VPR;
#pragma unroll
  for (int it = 0; it < (TOT + BLOCK - 1) / BLOCK; ++it) {
    int idx = it * BLOCK + threadIdx.x;
    if (idx < TOT) {
      int r = idx / VPR, c8 = idx % VPR;
      bf16x8 v = *(const bf16x8*)(src + r * src_row_stride + c8 * 8);
      *(bf16x8*)((char*)dst + swz(r, c8 * 8, ATT_D * 2)) = v;
    }
  }
}

// stage a [ROWS][128] global tile TRANSPOSED into a [128][ROWS] swizzled
// LDS tile (ROWS = 32 or 64; rowb = ROWS*2 bytes)
template <int ROWS, int BLOCK>
DEV void stage_tr(short* dst, const short* src, long long src_row_stride) {
  constexpr int VPR = ATT_D / 8;
  constexpr int TOT = ROWS * VPR;
#pragma unroll
  for (int it = 0; it < (TOT + BLOCK - 1) / BLOCK; ++it) {
    int idx = it * BLOCK + threadIdx.x;
    if (idx < TOT) {
      int r = idx / VPR, c8 = idx % VPR;
      bf16x8 v = *(const bf16x8*)(src + r * src_row_stride + c8 * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *(short*)((char*)dst + swz(c8 * 8 + j, r, ROWS * 2)) = v[j];
    }
  }
}

// intra-wave LDS exchange fence: the P/dS round-trips move data between
// lanes of the SAME wave; DS ops issue in order per wave, so a full block
// barrier is overkill — we only need the writes drained and the compiler
// forbidden from reordering the reads above the writes (per-lane alias
// analysis would otherwise allow it).
DEV void wave_lds_fence() {
  __builtin_amdgcn_s_waitcnt(0);
  __builtin_amdgcn_wave_barrier();
}

// ===================== BACKWARD: delta preprocess ==========================
// delta[row] = rowsum(dO[row] * O[row]); one wave per row, 2 elems/lane.
__global__ void attn_bwd_delta_kernel(const short* __restrict__ dout,
                                      const short* __restrict__ o,
                                      float* __restrict__ delta,
                                      long long rows) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long long row = (long long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= rows) return;
  const short* dr = dout + row * ATT_D;
  const short* orow = o + row * ATT_D;
  float s = bf2f(dr[lane]) * bf2f(orow[lane])
            + bf2f(dr[lane + 64]) * bf2f(orow[lane + 64]);
  s = wave_sum(s);
  if (lane == 0) delta[row] = s;
}
