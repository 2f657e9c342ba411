// Flash-decode for gfx950: one query row per (b, h) against the KV
// cache, split-K across the cache length.
//
// Decode is MEMORY-bound (stream L x 128 x 2 tensors of bf16 per head);
// the design maximizes streaming bandwidth and chip fill, not MFMA:
//  * grid (splits, H, B): split-K gives >> 256 workgroups even at
//    batch 1 (B*H alone cannot fill 256 CUs);
//  * 4 waves per block, each wave owns 64-row tiles of its split
//    (row-per-lane scores: lane r streams K[row r] as 16-byte vec8
//    loads — 64 consecutive rows per wave = perfectly coalesced);
//  * online softmax per wave (wave_max/wave_sum over the 64 scores),
//    V accumulated d-per-lane (lane owns 2 of the 128 columns, p_r
//    broadcast by shfl) so the [row -> column] transpose never touches
//    memory;
//  * per-wave partials merge through LDS, per-split partials (o, m, l)
//    merge in a tiny second kernel (attn_decode_merge_kernel).
//
// The cache tensors keep their FULL allocation stride (Lmax): no
// slicing copies; L marks the valid prefix.

#include "common.h"

#define DEC_D 128

// partial per (split, b, h): o [splits,B,H,128] fp32, ml [splits,B,H,2]
__global__ __launch_bounds__(256) void attn_decode_partial_kernel(
    const short* __restrict__ q,    // [B, H, 1, 128]
    const short* __restrict__ kc,   // [B, Hkv, Lmax, 128]
    const short* __restrict__ vc,
    const int* __restrict__ lengths,  // [B] valid prefix per sequence
    float* __restrict__ o_part, float* __restrict__ ml_part,
    int B, int H, int Hkv, int Lmax, int splits, float scale) {
  const int sp = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (H / Hkv);
  const int L = lengths[b];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);

  __shared__ float q_s[DEC_D];
  __shared__ float red_m[4], red_l[4];
  __shared__ float red_o[4][DEC_D];

  // stage the query row as fp32
  if (threadIdx.x < DEC_D)
    q_s[threadIdx.x] = bf2f(q[((long long)b * H + h) * DEC_D
                              + threadIdx.x]);
  __syncthreads();

  const int chunk = (L + splits - 1) / splits;
  const int r0 = sp * chunk;
  const int r1 = min(L, r0 + chunk);
  const long long kvoff = ((long long)b * Hkv + hk) * Lmax * DEC_D;

  float m_run = -INFINITY, l_run = 0.f;
  float o_acc[2] = {0.f, 0.f};  // lane owns d = 2*lane, 2*lane+1

  for (int base = r0 + wid * WAVE; base < r1; base += 4 * WAVE) {
    const int row = base + lane;
    const bool valid = row < r1;
    // score: dot(q, K[row]) — vec8 streaming loads
    float s = -INFINITY;
    if (valid) {
      const short* krow = kc + kvoff + (long long)row * DEC_D;
      float acc = 0.f;
#pragma unroll
      for (int d8 = 0; d8 < DEC_D / 8; ++d8) {
        bf16x8 kv8 = *(const bf16x8*)(krow + d8 * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc += q_s[d8 * 8 + j] * bf2f(kv8[j]);
      }
      s = acc * scale;
    }
    const float m_new = fmaxf(m_run, wave_max(s));
    const float rescale = __expf(m_run - m_new);
    const float p = valid ? __expf(s - m_new) : 0.f;
    l_run = l_run * rescale + wave_sum(p);
    o_acc[0] *= rescale;
    o_acc[1] *= rescale;
    m_run = m_new;
    // V accumulate: lane owns 2 d-columns; p_r broadcast per row
    const int nrows = min(WAVE, r1 - base);
    for (int r = 0; r < nrows; ++r) {
      const float pr = __shfl(p, r, WAVE);
      if (pr != 0.f) {
        const short* vrow = vc + kvoff + (long long)(base + r) * DEC_D;
        o_acc[0] += pr * bf2f(vrow[2 * lane]);
        o_acc[1] += pr * bf2f(vrow[2 * lane + 1]);
      }
    }
  }

  // merge the 4 waves through LDS
  if (lane == 0) {
    red_m[wid] = m_run;
    red_l[wid] = l_run;
  }
  red_o[wid][2 * lane] = o_acc[0];
  red_o[wid][2 * lane + 1] = o_acc[1];
  __syncthreads();
  if (wid == 0) {
    float m_blk = -INFINITY;
#pragma unroll
    for (int w = 0; w < 4; ++w) m_blk = fmaxf(m_blk, red_m[w]);
    float l_blk = 0.f;
    float o0 = 0.f, o1 = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float f = (red_m[w] == -INFINITY) ? 0.f
                      : __expf(red_m[w] - m_blk);
      l_blk += f * red_l[w];
      o0 += f * red_o[w][2 * lane];
      o1 += f * red_o[w][2 * lane + 1];
    }
    const long long po =
        (((long long)sp * B + b) * H + h) * DEC_D;
    o_part[po + 2 * lane] = o0;
    o_part[po + 2 * lane + 1] = o1;
    if (lane == 0) {
      const long long pm = (((long long)sp * B + b) * H + h) * 2;
      ml_part[pm] = m_blk;
      ml_part[pm + 1] = l_blk;
    }
  }
}

// merge the split partials: one wave per (b, h), lane owns 2 d-columns
__global__ __launch_bounds__(64) void attn_decode_merge_kernel(
    const float* __restrict__ o_part, const float* __restrict__ ml_part,
    short* __restrict__ o, int B, int H, int splits) {
  const int h = blockIdx.x % H;
  const int b = blockIdx.x / H;
  const int lane = threadIdx.x & (WAVE - 1);
  float m_all = -INFINITY;
  for (int sp = 0; sp < splits; ++sp)
    m_all = fmaxf(m_all,
                  ml_part[(((long long)sp * B + b) * H + h) * 2]);
  float l_all = 0.f, o0 = 0.f, o1 = 0.f;
  for (int sp = 0; sp < splits; ++sp) {
    const long long pm = (((long long)sp * B + b) * H + h) * 2;
    const float m_s = ml_part[pm];
    const float f = (m_s == -INFINITY) ? 0.f : __expf(m_s - m_all);
    l_all += f * ml_part[pm + 1];
    const long long po = (((long long)sp * B + b) * H + h) * DEC_D;
    o0 += f * o_part[po + 2 * lane];
    o1 += f * o_part[po + 2 * lane + 1];
  }
  // inactive slots (length 0) produce l_all == 0: write 0, not NaN
  const float inv = l_all > 0.f ? 1.f / l_all : 0.f;
  const long long oo = ((long long)b * H + h) * DEC_D;
  o[oo + 2 * lane] = f2bf(o0 * inv);
  o[oo + 2 * lane + 1] = f2bf(o1 * inv);
}
