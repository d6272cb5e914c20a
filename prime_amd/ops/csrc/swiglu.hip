// SwiGLU activation for gfx950: out = silu(gate) * up, with gate/up packed
// as one [R, 2I] tensor (gate = [:, :I], up = [:, I:]) so the preceding
// hipBLASLt GEMM produces both halves in one call.
//
// Memory-bound; bf16x8 vectorized; backward recomputes silu from the saved
// input (no extra activation storage).
#include "common.h"

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.f / (1.f + __expf(-x));
}

__global__ void swiglu_fwd_kernel(const bf16* __restrict__ gu,
                                  bf16* __restrict__ out, int64_t R, int I) {
  const int iv = I / 8;
  const int64_t total = R * iv;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = idx / iv;
    const int i = (int)(idx - r * iv) * 8;
    const bf16x8 g = *reinterpret_cast<const bf16x8*>(gu + r * 2 * I + i);
    const bf16x8 u = *reinterpret_cast<const bf16x8*>(gu + r * 2 * I + I + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf2f(g.v[j]);
      o.v[j] = f2bf(gf * sigmoidf_(gf) * bf2f(u.v[j]));
    }
    *reinterpret_cast<bf16x8*>(out + r * I + i) = o;
  }
}

__global__ void swiglu_bwd_kernel(const bf16* __restrict__ dout,
                                  const bf16* __restrict__ gu,
                                  bf16* __restrict__ dgu, int64_t R, int I) {
  const int iv = I / 8;
  const int64_t total = R * iv;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = idx / iv;
    const int i = (int)(idx - r * iv) * 8;
    const bf16x8 g = *reinterpret_cast<const bf16x8*>(gu + r * 2 * I + i);
    const bf16x8 u = *reinterpret_cast<const bf16x8*>(gu + r * 2 * I + I + i);
    const bf16x8 d = *reinterpret_cast<const bf16x8*>(dout + r * I + i);
    bf16x8 dg, du;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf2f(g.v[j]);
      const float sg = sigmoidf_(gf);
      const float silu = gf * sg;
      const float df = bf2f(d.v[j]);
      dg.v[j] = f2bf(df * bf2f(u.v[j]) * sg * (1.f + gf * (1.f - sg)));
      du.v[j] = f2bf(df * silu);
    }
    *reinterpret_cast<bf16x8*>(dgu + r * 2 * I + i) = dg;
    *reinterpret_cast<bf16x8*>(dgu + r * 2 * I + I + i) = du;
  }
}

PRIME_API int prime_swiglu_fwd(hipStream_t stream, const void* gu, void* out,
                               int64_t R, int64_t I) {
  if (I % 8 != 0) return hipErrorInvalidValue;
  int grid = prime_grid(R * (I / 8), 256);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)gu, (bf16*)out, R, (int)I);
  return (int)hipGetLastError();
}

PRIME_API int prime_swiglu_bwd(hipStream_t stream, const void* dout,
                               const void* gu, void* dgu, int64_t R, int64_t I) {
  if (I % 8 != 0) return hipErrorInvalidValue;
  int grid = prime_grid(R * (I / 8), 256);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)dout, (const bf16*)gu, (bf16*)dgu, R, (int)I);
  return (int)hipGetLastError();
}
