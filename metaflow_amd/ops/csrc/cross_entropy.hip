// Fused cross-entropy over a [N, V] bf16 logit matrix (V up to ~128k).
//
// Forward computes per-row loss AND writes dlogits = (softmax - onehot) *
// grad_scale in the same kernel family, so the huge logit tensor is read
// twice and written once total — the eager path (softmax + nll + backward)
// reads/writes it ~5x. Block-per-row, online max+sum, bf16x8 loads.

#include "common.h"

template <int BLOCK>
__global__ void cross_entropy_fwd_kernel(const short* __restrict__ logits,
                                         const long long* __restrict__ tgt,
                                         float* __restrict__ loss,
                                         float* __restrict__ row_max,
                                         float* __restrict__ row_lse,
                                         long long N, int V,
                                         long long ignore_index) {
  __shared__ float scratch[BLOCK / WAVE];
  for (long long row = blockIdx.x; row < N; row += gridDim.x) {
    const short* lr = logits + row * (long long)V;
    const long long t = tgt[row];

    float mx = -INFINITY;
    const int VV = V / 8;
    const bf16x8* lv = (const bf16x8*)lr;
    for (int i = threadIdx.x; i < VV; i += BLOCK) {
      bf16x8 x = lv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) mx = fmaxf(mx, bf2f(x[j]));
    }
    for (int i = VV * 8 + threadIdx.x; i < V; i += BLOCK)
      mx = fmaxf(mx, bf2f(lr[i]));
    mx = block_max<BLOCK>(mx, scratch);

    float s = 0.f;
    for (int i = threadIdx.x; i < VV; i += BLOCK) {
      bf16x8 x = lv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) s += __expf(bf2f(x[j]) - mx);
    }
    for (int i = VV * 8 + threadIdx.x; i < V; i += BLOCK)
      s += __expf(bf2f(lr[i]) - mx);
    s = block_sum<BLOCK>(s, scratch);
    const float lse = __logf(s) + mx;

    if (threadIdx.x == 0) {
      row_max[row] = mx;
      row_lse[row] = lse;
      if (t == ignore_index)
        loss[row] = 0.f;
      else
        loss[row] = lse - bf2f(lr[t]);
    }
    __syncthreads();
  }
}

template <int BLOCK>
__global__ void cross_entropy_bwd_kernel(const short* __restrict__ logits,
                                         const long long* __restrict__ tgt,
                                         const float* __restrict__ row_lse,
                                         const float* __restrict__ dloss,
                                         short* __restrict__ dlogits,
                                         long long N, int V,
                                         long long ignore_index) {
  for (long long row = blockIdx.x; row < N; row += gridDim.x) {
    const short* lr = logits + row * (long long)V;
    short* dr = dlogits + row * (long long)V;
    const long long t = tgt[row];
    const float lse = row_lse[row];
    const float scale = (t == ignore_index) ? 0.f : dloss[row];

    const int VV = V / 8;
    const bf16x8* lv = (const bf16x8*)lr;
    bf16x8* dv = (bf16x8*)dr;
    for (int i = threadIdx.x; i < VV; i += BLOCK) {
      bf16x8 x = lv[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const long long col = (long long)i * 8 + j;
        float p = __expf(bf2f(x[j]) - lse);
        o[j] = f2bf((p - (col == t ? 1.f : 0.f)) * scale);
      }
      dv[i] = o;
    }
    for (int i = VV * 8 + threadIdx.x; i < V; i += BLOCK) {
      float p = __expf(bf2f(lr[i]) - lse);
      dr[i] = f2bf((p - ((long long)i == t ? 1.f : 0.f)) * scale);
    }
    __syncthreads();
  }
}
