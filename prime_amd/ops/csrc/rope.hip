// Rotary position embedding (NeoX half-split / Llama "rotate_half") for
// gfx950. Elementwise + table-lookup: cos/sin are precomputed on HOST
// (guide App. B: on-device trig turns memory-bound into VALU-bound) and
// passed as fp32 tables [S, D/2].
//
// Tensor layout: [B, S, H, D] bf16, treated as rows of length D where
// row -> position = (row / H) % S. Forward and backward share a kernel
// (backward is rotation by -theta: sign = -1).
#include "common.h"

__global__ void rope_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t, int64_t nrows,
                            int H, int S, int D, float sign,
                            int64_t pos_offset,
                            const int* __restrict__ pos_dev) {
  if (pos_dev) pos_offset = *pos_dev;  // hipGraph decode: dynamic position
  const int half = D / 2;  // multiple of 4 (head_dim 64/128)
  const int hv = half / 4;
  const int64_t total = nrows * hv;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / hv;
    const int i4 = (int)(idx - row * hv) * 4;  // pair index within half
    const int64_t s = (row / H) % S + pos_offset;
    const bf16x4 x1 = *reinterpret_cast<const bf16x4*>(x + row * D + i4);
    const bf16x4 x2 = *reinterpret_cast<const bf16x4*>(x + row * D + half + i4);
    const f32x4v c = *reinterpret_cast<const f32x4v*>(cos_t + s * half + i4);
    const f32x4v sn = *reinterpret_cast<const f32x4v*>(sin_t + s * half + i4);
    bf16x4 y1, y2;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float a = bf2f(x1.v[j]), b = bf2f(x2.v[j]);
      const float sj = sign * sn.v[j];
      y1.v[j] = f2bf(a * c.v[j] - b * sj);
      y2.v[j] = f2bf(b * c.v[j] + a * sj);
    }
    *reinterpret_cast<bf16x4*>(y + row * D + i4) = y1;
    *reinterpret_cast<bf16x4*>(y + row * D + half + i4) = y2;
  }
}

PRIME_API int prime_rope(hipStream_t stream, const void* x, void* y,
                         const void* cos_t, const void* sin_t, int64_t nrows,
                         int64_t H, int64_t S, int64_t D, int backward,
                         int64_t pos_offset, const void* pos_dev) {
  if (D % 8 != 0) return hipErrorInvalidValue;
  int64_t total = nrows * (D / 8);
  int grid = prime_grid(total, 256);
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)x, (bf16*)y, (const float*)cos_t,
                     (const float*)sin_t, nrows, (int)H, (int)S, (int)D,
                     backward ? -1.f : 1.f, pos_offset, (const int*)pos_dev);
  return (int)hipGetLastError();
}
