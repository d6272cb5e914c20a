// RMSNorm forward/backward for MI355X (gfx950).
//
// Memory-bound: vectorized bf16x8 loads (16 B/lane), fp32 accumulation,
// one 256-thread block per row (grid-stride over rows). Backward
// accumulates dw into an LDS fp32 buffer per block, then one atomicAdd
// pass per block (device-scope atomics are XCD-safe, guide §6 G12).
//
// Mirrors the capability of the reference framework's fused norm path
// (SURVEY.md §B3); numerics checked against PyTorch fp32 reference in
// tests/test_ops_gpu.py.
#include "common.h"

// y[r,:] = x[r,:] * rsqrt(mean(x^2)+eps) * w ; rstd[r] saved for bwd.
__global__ void rmsnorm_fwd_kernel(const bf16* __restrict__ x,
                                   const bf16* __restrict__ w,
                                   bf16* __restrict__ y,
                                   float* __restrict__ rstd,
                                   int64_t R, int D, float eps) {
  __shared__ float scratch[16];
  const int tid = threadIdx.x;
  const int nthr = blockDim.x;
  const int dvec = D / 8;
  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const bf16x8* xr = reinterpret_cast<const bf16x8*>(x + r * D);
    float ss = 0.f;
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v.v[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum(ss, scratch);
    float rs = rsqrtf(ss / (float)D + eps);
    if (tid == 0) rstd[r] = rs;
    bf16x8* yr = reinterpret_cast<bf16x8*>(y + r * D);
    const bf16x8* wv = reinterpret_cast<const bf16x8*>(w);
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 v = xr[i], wj = wv[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o.v[j] = f2bf(bf2f(v.v[j]) * rs * bf2f(wj.v[j]));
      yr[i] = o;
    }
  }
}

// dx = rstd * (dy*w - xhat * mean(dy*w*xhat)), xhat = x*rstd
// dw += sum_r dy * xhat   (fp32 accumulation in LDS, then global atomics)
__global__ void rmsnorm_bwd_kernel(const bf16* __restrict__ dy,
                                   const bf16* __restrict__ x,
                                   const bf16* __restrict__ w,
                                   const float* __restrict__ rstd,
                                   bf16* __restrict__ dx,
                                   float* __restrict__ dw,
                                   int64_t R, int D, float eps) {
  extern __shared__ float lds[];        // [D] dw accumulator + 16 scratch
  float* dw_loc = lds;
  float* scratch = lds + D;
  const int tid = threadIdx.x;
  const int nthr = blockDim.x;
  const int dvec = D / 8;
  for (int i = tid; i < D; i += nthr) dw_loc[i] = 0.f;
  __syncthreads();

  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const bf16x8* dyr = reinterpret_cast<const bf16x8*>(dy + r * D);
    const bf16x8* xr = reinterpret_cast<const bf16x8*>(x + r * D);
    const bf16x8* wv = reinterpret_cast<const bf16x8*>(w);
    const float rs = rstd[r];
    // pass 1: dot = sum(dy*w*xhat)
    float dot = 0.f;
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 d = dyr[i], xv = xr[i], wj = wv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += bf2f(d.v[j]) * bf2f(wj.v[j]) * bf2f(xv.v[j]) * rs;
    }
    dot = block_reduce_sum(dot, scratch);
    const float mean_dot = dot / (float)D;
    bf16x8* dxr = reinterpret_cast<bf16x8*>(dx + r * D);
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 d = dyr[i], xv = xr[i], wj = wv[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xhat = bf2f(xv.v[j]) * rs;
        float dyw = bf2f(d.v[j]) * bf2f(wj.v[j]);
        o.v[j] = f2bf(rs * (dyw - xhat * mean_dot));
        dw_loc[i * 8 + j] += bf2f(d.v[j]) * xhat;
      }
      dxr[i] = o;
    }
    __syncthreads();  // dw_loc writes race-free across rows (same owners), but
                      // keep row boundary ordered with the reduction scratch
  }
  for (int i = tid; i < D; i += nthr)
    if (dw_loc[i] != 0.f) atomicAdd(&dw[i], dw_loc[i]);
}

PRIME_API int prime_rmsnorm_fwd(hipStream_t stream, const void* x, const void* w,
                                void* y, void* rstd, int64_t R, int64_t D,
                                double eps) {
  if (D % 8 != 0) return hipErrorInvalidValue;
  int grid = prime_grid(R, 1);
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)x, (const bf16*)w, (bf16*)y, (float*)rstd, R,
                     (int)D, (float)eps);
  return (int)hipGetLastError();
}

PRIME_API int prime_rmsnorm_bwd(hipStream_t stream, const void* dy, const void* x,
                                const void* w, const void* rstd, void* dx,
                                void* dw, int64_t R, int64_t D, double eps) {
  if (D % 8 != 0) return hipErrorInvalidValue;
  int grid = prime_grid(R, 1);
  if (grid > 1024) grid = 1024;  // bound atomic traffic on dw
  size_t lds = (size_t)(D + 16) * sizeof(float);
  if (lds > 160 * 1024 - 1024) return hipErrorInvalidValue;
  hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(grid), dim3(256), lds, stream,
                     (const bf16*)dy, (const bf16*)x, (const bf16*)w,
                     (const float*)rstd, (bf16*)dx, (float*)dw, R, (int)D,
                     (float)eps);
  return (int)hipGetLastError();
}
