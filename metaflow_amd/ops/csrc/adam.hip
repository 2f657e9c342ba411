// Fused AdamW for bf16 training on gfx950.
//
// Params/grads bf16, moments fp32, optional fp32 master weights. One pass:
// read g, m, v (+master or p), update, write p (+master), all vectorized.
// Memory-bound: ~26 B/element traffic, target HBM ceiling.

#include "common.h"

__global__ void adamw_kernel(short* __restrict__ p,          // bf16 params
                             const short* __restrict__ g,    // bf16 grads
                             float* __restrict__ m,
                             float* __restrict__ v,
                             float* __restrict__ master,     // may be null
                             long long n,
                             float lr, float beta1, float beta2, float eps,
                             float weight_decay, float bias_c1,
                             float bias_c2, float grad_scale) {
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (long long i0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < n; i0 += stride) {
    // tail-safe scalar quad (n is padded to multiple of 4 by the wrapper)
    bf16x4 gp = *(const bf16x4*)(g + i0);
    f32x4 mv = *(f32x4*)(m + i0);
    f32x4 vv = *(f32x4*)(v + i0);
    f32x4 pv;
    if (master) {
      pv = *(f32x4*)(master + i0);
    } else {
      bf16x4 pb = *(bf16x4*)(p + i0);
#pragma unroll
      for (int j = 0; j < 4; ++j) pv[j] = bf2f(pb[j]);
    }
    bf16x4 pout;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = bf2f(gp[j]) * grad_scale;
      float mf = beta1 * mv[j] + (1.f - beta1) * gf;
      float vf = beta2 * vv[j] + (1.f - beta2) * gf * gf;
      float mhat = mf * bias_c1;
      float vhat = vf * bias_c2;
      float pf = pv[j];
      pf -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * pf);
      mv[j] = mf;
      vv[j] = vf;
      pv[j] = pf;
      pout[j] = f2bf(pf);
    }
    *(f32x4*)(m + i0) = mv;
    *(f32x4*)(v + i0) = vv;
    if (master) *(f32x4*)(master + i0) = pv;
    *(bf16x4*)(p + i0) = pout;
  }
}

// fp32 variant (for fp32 params, e.g. norms kept in fp32)
__global__ void adamw_f32_kernel(float* __restrict__ p,
                                 const float* __restrict__ g,
                                 float* __restrict__ m,
                                 float* __restrict__ v,
                                 long long n,
                                 float lr, float beta1, float beta2,
                                 float eps, float weight_decay,
                                 float bias_c1, float bias_c2,
                                 float grad_scale) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    float gf = g[i] * grad_scale;
    float mf = beta1 * m[i] + (1.f - beta1) * gf;
    float vf = beta2 * v[i] + (1.f - beta2) * gf * gf;
    float pf = p[i];
    pf -= lr * (mf * bias_c1 / (sqrtf(vf * bias_c2) + eps)
                + weight_decay * pf);
    m[i] = mf;
    v[i] = vf;
    p[i] = pf;
  }
}
