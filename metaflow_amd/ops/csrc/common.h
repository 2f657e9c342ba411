// Common helpers for metaflow_amd CDNA4 (gfx950) kernels.
// Wave = 64 lanes; LDS = 160 KiB/CU, 32 banks x 4B; vectorize bf16 as
// short4/short8 (guide G13); grid-stride memory-bound kernels capped at
// ~2048 blocks (guide G11).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) short bf16x8;  // 4 VGPRs
typedef __attribute__((ext_vector_type(4))) short bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;

DEV float bf2f(short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

DEV short f2bf(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  // round-to-nearest-even
  unsigned int lsb = (c.i >> 16) & 1;
  c.i += 0x7fffu + lsb;
  return (short)(c.i >> 16);
}

// wave-wide reductions (64 lanes)
DEV float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

DEV float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// block-wide reduce across waves through LDS; BLOCK must be multiple of 64.
template <int BLOCK>
DEV float block_sum(float v, float* scratch /* BLOCK/64 floats */) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  v = (threadIdx.x < NW) ? scratch[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1)
      v += __shfl_xor(v, off, WAVE);
    if (lane == 0) scratch[0] = v;
  }
  __syncthreads();
  float r = scratch[0];
  __syncthreads();
  return r;
}

template <int BLOCK>
DEV float block_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  v = (threadIdx.x < NW) ? scratch[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1)
      v = fmaxf(v, __shfl_xor(v, off, WAVE));
    if (lane == 0) scratch[0] = v;
  }
  __syncthreads();
  float r = scratch[0];
  __syncthreads();
  return r;
}

#define HIP_CHECK_KERNEL()                                    \
  do {                                                        \
    hipError_t e = hipGetLastError();                         \
    if (e != hipSuccess) {                                    \
      TORCH_CHECK(false, "HIP kernel launch failed: ",        \
                  hipGetErrorString(e));                      \
    }                                                         \
  } while (0)
