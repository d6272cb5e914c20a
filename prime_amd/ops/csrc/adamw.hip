// Fused AdamW for gfx950: one pass over {bf16 grad, fp32 master param,
// fp32 m, fp32 v} updating master + exp-avgs and emitting the bf16 model
// param — 5 tensors touched once instead of PyTorch's ~10 kernel launches.
//
// Decoupled weight decay (AdamW); bias correction folded into host-side
// scalars c1 = 1/(1-beta1^t), c2 = 1/(1-beta2^t). Memory-bound: float4 /
// bf16x4 vectorized, grid-stride.
#include "common.h"

__global__ void adamw_kernel(float* __restrict__ p32, bf16* __restrict__ p16,
                             const bf16* __restrict__ g, float* __restrict__ m,
                             float* __restrict__ v,
                             const float* __restrict__ gscale, int64_t N,
                             float lr, float beta1, float beta2, float eps,
                             float wd, float c1, float c2) {
  // optional fused gradient clip: gscale points at a 1-element device
  // scalar (min(1, max_norm/||g||)) computed by grad_sqnorm + host math —
  // avoids a separate 21 GB bf16 read+write pass over the flat grad
  const float gs = gscale ? *gscale : 1.f;
  const int64_t nv = N / 4;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < nv;
       idx += (int64_t)gridDim.x * blockDim.x) {
    f32x4v pv = reinterpret_cast<f32x4v*>(p32)[idx];
    f32x4v mv = reinterpret_cast<f32x4v*>(m)[idx];
    f32x4v vv = reinterpret_cast<f32x4v*>(v)[idx];
    const bf16x4 gv = reinterpret_cast<const bf16x4*>(g)[idx];
    bf16x4 pb;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float gf = bf2f(gv.v[j]) * gs;
      mv.v[j] = beta1 * mv.v[j] + (1.f - beta1) * gf;
      vv.v[j] = beta2 * vv.v[j] + (1.f - beta2) * gf * gf;
      const float mhat = mv.v[j] * c1;
      const float vhat = vv.v[j] * c2;
      pv.v[j] -= lr * (mhat / (sqrtf(vhat) + eps) + wd * pv.v[j]);
      pb.v[j] = f2bf(pv.v[j]);
    }
    reinterpret_cast<f32x4v*>(p32)[idx] = pv;
    reinterpret_cast<f32x4v*>(m)[idx] = mv;
    reinterpret_cast<f32x4v*>(v)[idx] = vv;
    reinterpret_cast<bf16x4*>(p16)[idx] = pb;
  }
  // scalar tail
  const int64_t tail = nv * 4;
  for (int64_t i = tail + blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float gf = bf2f(g[i]) * gs;
    m[i] = beta1 * m[i] + (1.f - beta1) * gf;
    v[i] = beta2 * v[i] + (1.f - beta2) * gf * gf;
    p32[i] -= lr * ((m[i] * c1) / (sqrtf(v[i] * c2) + eps) + wd * p32[i]);
    p16[i] = f2bf(p32[i]);
  }
}

PRIME_API int prime_adamw(hipStream_t stream, void* p32, void* p16,
                          const void* g, void* m, void* v, const void* gscale,
                          int64_t N, double lr, double beta1, double beta2,
                          double eps, double wd, int64_t step) {
  const float b1 = (float)beta1, b2 = (float)beta2;
  const float c1 = 1.f / (1.f - powf(b1, (float)step));
  const float c2 = 1.f / (1.f - powf(b2, (float)step));
  int grid = prime_grid(N / 4 + 1, 256);
  hipLaunchKernelGGL(adamw_kernel, dim3(grid), dim3(256), 0, stream,
                     (float*)p32, (bf16*)p16, (const bf16*)g, (float*)m,
                     (float*)v, (const float*)gscale, N, (float)lr, b1, b2,
                     (float)eps, (float)wd, c1, c2);
  return (int)hipGetLastError();
}

// single-pass squared L2 norm of a bf16 buffer (fp32 accumulation)
__global__ void sqnorm_kernel(const bf16* __restrict__ g, float* __restrict__ out,
                              int64_t N) {
  __shared__ float scratch[16];
  float acc = 0.f;
  const int64_t nv = N / 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    const bf16x8 v = reinterpret_cast<const bf16x8*>(g)[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = bf2f(v.v[j]);
      acc += f * f;
    }
  }
  for (int64_t i = nv * 8 + blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float f = bf2f(g[i]);
    acc += f * f;
  }
  acc = block_reduce_sum(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

PRIME_API int prime_grad_sqnorm(hipStream_t stream, const void* g, void* out,
                                int64_t N) {
  int grid = prime_grid(N / 8 + 1, 256);
  hipLaunchKernelGGL(sqnorm_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)g, (float*)out, N);
  return (int)hipGetLastError();
}
