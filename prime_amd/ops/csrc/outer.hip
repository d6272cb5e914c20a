// DiLoCo outer-step kernels for gfx950:
//   - pseudo-gradient:  delta = theta_outer(fp32) - theta_inner(bf16)
//   - blockwise int8 quantize / dequantize(+add) for the ring all-reduce
//     (per-1024-element absmax scale; comm volume = 1/4 of fp32 + 0.4% scales)
//   - fused Nesterov outer update that consumes the averaged pseudo-grad,
//     updates outer fp32 weights + momentum, and emits the new bf16 inner
//     params in the same pass (guide §B3: fuse dequant consumer with update).
#include "common.h"

#define QBLK 1024  // elements per int8 quantization block

// ---- pseudo-gradient ---------------------------------------------------
// delta = theta_outer - master32 (both fp32: the inner AdamW keeps an fp32
// master, so the pseudo-grad is exact rather than bf16-rounded)
__global__ void pseudograd_kernel(const float* __restrict__ outer,
                                  const float* __restrict__ inner,
                                  float* __restrict__ delta, int64_t N) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x)
    delta[i] = outer[i] - inner[i];
}

// ---- int8 block quant --------------------------------------------------
// one 256-thread block per QBLK chunk (4 elems/thread), absmax via block
// reduction; scale = absmax/127; q = rint(x/scale).
__global__ void quant_int8_kernel(const float* __restrict__ x,
                                  int8_t* __restrict__ q,
                                  float* __restrict__ scales, int64_t nblk,
                                  int64_t N) {
  __shared__ float scratch[16];
  for (int64_t b = blockIdx.x; b < nblk; b += gridDim.x) {
    const int64_t base = b * QBLK;
    const int64_t lim = min((int64_t)QBLK, N - base);
    float amax = 0.f;
    float vals[4];
    const int i0 = threadIdx.x * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int64_t i = base + i0 + j;
      vals[j] = (i0 + j < lim) ? x[i] : 0.f;
      amax = fmaxf(amax, fabsf(vals[j]));
    }
    amax = block_reduce_max(amax, scratch);
    const float scale = amax / 127.f;
    const float inv = (scale > 0.f) ? 1.f / scale : 0.f;
    if (threadIdx.x == 0) scales[b] = scale;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      if (i0 + j < lim)
        q[base + i0 + j] = (int8_t)__float2int_rn(
            fminf(127.f, fmaxf(-127.f, vals[j] * inv)));
  }
}

// dst (+)= q * scale
__global__ void dequant_int8_kernel(const int8_t* __restrict__ q,
                                    const float* __restrict__ scales,
                                    float* __restrict__ dst, int64_t N,
                                    int accumulate) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float v = (float)q[i] * scales[i / QBLK];
    dst[i] = accumulate ? dst[i] + v : v;
  }
}

// ---- fused Nesterov outer update --------------------------------------
// delta_avg: averaged pseudo-grad (direction outer - inner_avg).
// buf = mu*buf + delta ; theta -= lr*(delta + mu*buf); then the inner state
// restarts from the new outer point: master32 = theta, inner16 = bf16(theta).
__global__ void nesterov_kernel(float* __restrict__ theta,
                                float* __restrict__ master32,
                                bf16* __restrict__ inner16,
                                float* __restrict__ buf,
                                const float* __restrict__ delta, int64_t N,
                                float lr, float mu) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float d = delta[i];
    const float b = mu * buf[i] + d;
    buf[i] = b;
    const float t = theta[i] - lr * (d + mu * b);
    theta[i] = t;
    master32[i] = t;
    inner16[i] = f2bf(t);
  }
}

PRIME_API int prime_pseudograd(hipStream_t stream, const void* outer,
                               const void* inner, void* delta, int64_t N) {
  int grid = prime_grid(N, 256);
  hipLaunchKernelGGL(pseudograd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const float*)outer, (const float*)inner, (float*)delta,
                     N);
  return (int)hipGetLastError();
}

PRIME_API int prime_quant_int8(hipStream_t stream, const void* x, void* q,
                               void* scales, int64_t N) {
  int64_t nblk = (N + QBLK - 1) / QBLK;
  int grid = prime_grid(nblk, 1);
  hipLaunchKernelGGL(quant_int8_kernel, dim3(grid), dim3(256), 0, stream,
                     (const float*)x, (int8_t*)q, (float*)scales, nblk, N);
  return (int)hipGetLastError();
}

PRIME_API int prime_dequant_int8(hipStream_t stream, const void* q,
                                 const void* scales, void* dst, int64_t N,
                                 int64_t accumulate) {
  int grid = prime_grid(N, 256);
  hipLaunchKernelGGL(dequant_int8_kernel, dim3(grid), dim3(256), 0, stream,
                     (const int8_t*)q, (const float*)scales, (float*)dst, N,
                     (int)accumulate);
  return (int)hipGetLastError();
}

PRIME_API int prime_nesterov_outer(hipStream_t stream, void* theta,
                                   void* master32, void* inner16, void* buf,
                                   const void* delta, int64_t N, double lr,
                                   double mu) {
  int grid = prime_grid(N, 256);
  hipLaunchKernelGGL(nesterov_kernel, dim3(grid), dim3(256), 0, stream,
                     (float*)theta, (float*)master32, (bf16*)inner16,
                     (float*)buf, (const float*)delta, N, (float)lr,
                     (float)mu);
  return (int)hipGetLastError();
}
