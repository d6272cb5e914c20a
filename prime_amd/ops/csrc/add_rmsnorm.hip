// Fused residual-add + RMSNorm for gfx950.
//
//   fwd: s = x + res (new residual stream, written once)
//        y = s * rsqrt(mean(s^2)+eps) * w
//   bwd: ds = ds_res + rmsnorm_dx(dy)   (residual grad fused in)
//
// Replaces the {eager add kernel, separate norm} pair per block side:
// one fewer full read+write pass of the hidden stream per fusion site
// (memory-bound; bf16x8 vectorized like rmsnorm.hip).
#include "common.h"

__global__ void add_rmsnorm_fwd_kernel(const bf16* __restrict__ x,
                                       const bf16* __restrict__ res,
                                       const bf16* __restrict__ w,
                                       bf16* __restrict__ s_out,
                                       bf16* __restrict__ y,
                                       float* __restrict__ rstd, int64_t R,
                                       int D, float eps) {
  __shared__ float scratch[16];
  const int tid = threadIdx.x;
  const int nthr = blockDim.x;
  const int dvec = D / 8;
  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const bf16x8* xr = reinterpret_cast<const bf16x8*>(x + r * D);
    const bf16x8* rr = res ? reinterpret_cast<const bf16x8*>(res + r * D) : nullptr;
    bf16x8* sr = reinterpret_cast<bf16x8*>(s_out + r * D);
    float ss = 0.f;
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 v = xr[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v.v[j]);
        if (rr) f += bf2f(rr[i].v[j]);
        o.v[j] = f2bf(f);
        ss += f * f;
      }
      sr[i] = o;
    }
    ss = block_reduce_sum(ss, scratch);
    const float rs = rsqrtf(ss / (float)D + eps);
    if (tid == 0) rstd[r] = rs;
    bf16x8* yr = reinterpret_cast<bf16x8*>(y + r * D);
    const bf16x8* wv = reinterpret_cast<const bf16x8*>(w);
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 v = sr[i], wj = wv[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o.v[j] = f2bf(bf2f(v.v[j]) * rs * bf2f(wj.v[j]));
      yr[i] = o;
    }
  }
}

// ds = ds_res + rstd*(dy*w - shat*mean(dy*w*shat)) ; dw += sum dy*shat
__global__ void add_rmsnorm_bwd_kernel(const bf16* __restrict__ dy,
                                       const bf16* __restrict__ ds_res,
                                       const bf16* __restrict__ s,
                                       const bf16* __restrict__ w,
                                       const float* __restrict__ rstd,
                                       bf16* __restrict__ ds_out,
                                       float* __restrict__ dw, int64_t R,
                                       int D, float eps) {
  extern __shared__ float lds[];
  float* dw_loc = lds;
  float* scratch = lds + D;
  const int tid = threadIdx.x;
  const int nthr = blockDim.x;
  const int dvec = D / 8;
  for (int i = tid; i < D; i += nthr) dw_loc[i] = 0.f;
  __syncthreads();

  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const bf16x8* dyr = reinterpret_cast<const bf16x8*>(dy + r * D);
    const bf16x8* srr = reinterpret_cast<const bf16x8*>(s + r * D);
    const bf16x8* dres = ds_res ? reinterpret_cast<const bf16x8*>(ds_res + r * D) : nullptr;
    const bf16x8* wv = reinterpret_cast<const bf16x8*>(w);
    const float rs = rstd[r];
    float dot = 0.f;
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 d = dyr[i], sv = srr[i], wj = wv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += bf2f(d.v[j]) * bf2f(wj.v[j]) * bf2f(sv.v[j]) * rs;
    }
    dot = block_reduce_sum(dot, scratch);
    const float mean_dot = dot / (float)D;
    bf16x8* dso = reinterpret_cast<bf16x8*>(ds_out + r * D);
    for (int i = tid; i < dvec; i += nthr) {
      bf16x8 d = dyr[i], sv = srr[i], wj = wv[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float shat = bf2f(sv.v[j]) * rs;
        const float dyw = bf2f(d.v[j]) * bf2f(wj.v[j]);
        float g = rs * (dyw - shat * mean_dot);
        if (dres) g += bf2f(dres[i].v[j]);
        o.v[j] = f2bf(g);
        dw_loc[i * 8 + j] += bf2f(d.v[j]) * shat;
      }
      dso[i] = o;
    }
    __syncthreads();
  }
  for (int i = tid; i < D; i += nthr)
    if (dw_loc[i] != 0.f) atomicAdd(&dw[i], dw_loc[i]);
}

PRIME_API int prime_add_rmsnorm_fwd(hipStream_t stream, const void* x,
                                    const void* res, const void* w, void* s_out,
                                    void* y, void* rstd, int64_t R, int64_t D,
                                    double eps) {
  if (D % 8 != 0) return hipErrorInvalidValue;
  int grid = prime_grid(R, 1);
  hipLaunchKernelGGL(add_rmsnorm_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)x, (const bf16*)res, (const bf16*)w,
                     (bf16*)s_out, (bf16*)y, (float*)rstd, R, (int)D,
                     (float)eps);
  return (int)hipGetLastError();
}

PRIME_API int prime_add_rmsnorm_bwd(hipStream_t stream, const void* dy,
                                    const void* ds_res, const void* s,
                                    const void* w, const void* rstd,
                                    void* ds_out, void* dw, int64_t R,
                                    int64_t D, double eps) {
  if (D % 8 != 0) return hipErrorInvalidValue;
  int grid = prime_grid(R, 1);
  if (grid > 1024) grid = 1024;
  size_t lds = (size_t)(D + 16) * sizeof(float);
  if (lds > 160 * 1024 - 1024) return hipErrorInvalidValue;
  hipLaunchKernelGGL(add_rmsnorm_bwd_kernel, dim3(grid), dim3(256), lds,
                     stream, (const bf16*)dy, (const bf16*)ds_res,
                     (const bf16*)s, (const bf16*)w, (const float*)rstd,
                     (bf16*)ds_out, (float*)dw, R, (int)D, (float)eps);
  return (int)hipGetLastError();
}
