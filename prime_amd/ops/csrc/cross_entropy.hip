// Fused cross-entropy fwd+bwd from bf16 logits for gfx950.
//
// One 256-thread block per row (grid-stride), online max/sumexp per thread
// then a block merge — a single read pass for the statistics and one
// read+write pass for the gradient, instead of PyTorch's softmax
// materialization (vocab 128k: saves a full [R,V] fp32 round-trip).
// Gradient is written as bf16 scaled by `grad_scale` (typically
// 1/n_valid_tokens); loss per row in fp32.
#include "common.h"

__global__ void ce_fwd_bwd_kernel(const bf16* __restrict__ logits,
                                  const int32_t* __restrict__ targets,
                                  float* __restrict__ loss,
                                  bf16* __restrict__ dlogits, int64_t R, int V,
                                  float grad_scale, int ignore_index,
                                  int compute_grad) {
  __shared__ float red_m[256];
  __shared__ float red_s[256];
  const int tid = threadIdx.x;
  const int nthr = blockDim.x;
  const int vv = V / 8;
  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const int tgt = targets[r];
    if (tgt == ignore_index) {
      if (tid == 0) loss[r] = 0.f;
      if (compute_grad) {
        bf16x8 z{};
#pragma unroll
        for (int j = 0; j < 8; ++j) z.v[j] = f2bf(0.f);
        bf16x8* dr = reinterpret_cast<bf16x8*>(dlogits + r * V);
        for (int i = tid; i < vv; i += nthr) dr[i] = z;
      }
      continue;
    }
    const bf16x8* zr = reinterpret_cast<const bf16x8*>(logits + r * V);
    float m = -INFINITY, s = 0.f;
    for (int i = tid; i < vv; i += nthr) {
      const bf16x8 z = zr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bf2f(z.v[j]);
        if (f > m) {
          s = s * __expf(m - f) + 1.f;
          m = f;
        } else {
          s += __expf(f - m);
        }
      }
    }
    red_m[tid] = m;
    red_s[tid] = s;
    __syncthreads();
    // tree-merge (m,s) pairs
    for (int stride = nthr / 2; stride > 0; stride >>= 1) {
      if (tid < stride) {
        const float m2 = red_m[tid + stride], s2 = red_s[tid + stride];
        const float mm = fmaxf(red_m[tid], m2);
        red_s[tid] = red_s[tid] * __expf(red_m[tid] - mm) + s2 * __expf(m2 - mm);
        red_m[tid] = mm;
      }
      __syncthreads();
    }
    const float M = red_m[0], S = red_s[0];
    __syncthreads();
    if (tid == 0) loss[r] = M + __logf(S) - bf2f(logits[r * V + tgt]);
    if (compute_grad) {
      const float invS = 1.f / S;
      bf16x8* dr = reinterpret_cast<bf16x8*>(dlogits + r * V);
      for (int i = tid; i < vv; i += nthr) {
        const bf16x8 z = zr[i];
        bf16x8 d;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int col = i * 8 + j;
          float p = __expf(bf2f(z.v[j]) - M) * invS;
          if (col == tgt) p -= 1.f;
          d.v[j] = f2bf(p * grad_scale);
        }
        dr[i] = d;
      }
    }
  }
}

PRIME_API int prime_cross_entropy(hipStream_t stream, const void* logits,
                                  const void* targets, void* loss,
                                  void* dlogits, int64_t R, int64_t V,
                                  double grad_scale, int64_t ignore_index,
                                  int64_t compute_grad) {
  if (V % 8 != 0) return hipErrorInvalidValue;
  int grid = prime_grid(R, 1);
  hipLaunchKernelGGL(ce_fwd_bwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)logits, (const int32_t*)targets, (float*)loss,
                     (bf16*)dlogits, R, (int)V, (float)grad_scale,
                     (int)ignore_index, (int)compute_grad);
  return (int)hipGetLastError();
}
