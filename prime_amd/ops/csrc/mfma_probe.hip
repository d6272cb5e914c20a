// Standalone MFMA layout probe: one wave computes C[16,16] = A[16,32]·B[32,16]
// with the fragment maps assumed by attention.hip. tests/test_ops_gpu.py
// checks it against torch.matmul with RANDOM ASYMMETRIC inputs (guide §3:
// symmetric inputs silently pass transposed layouts).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void mfma_probe_kernel(const bf16* __restrict__ A,
                                  const bf16* __restrict__ B,
                                  float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  // A[16][32] row-major: lane holds A[li][lg*8 + j]
  const short8 a = *reinterpret_cast<const short8*>(A + li * 32 + lg * 8);
  // B[32][16] row-major: lane holds B[lg*8 + j][li] -> strided gather
  short8 b;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    reinterpret_cast<short*>(&b)[j] =
        reinterpret_cast<const short*>(B)[(lg * 8 + j) * 16 + li];
  f32x4 c{0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  // C[16][16]: lane holds C[lg*4 + r][li]
#pragma unroll
  for (int r = 0; r < 4; ++r) C[(lg * 4 + r) * 16 + li] = c[r];
}

PRIME_API int prime_mfma_probe(hipStream_t stream, const void* A, const void* B,
                               void* C) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const bf16*)A, (const bf16*)B, (float*)C);
  return (int)hipGetLastError();
}

// ---- 32x32x16 probe (layout assumptions for the 32x32 attention path) ---
// A[32][16]: lane holds A[lane&31][(lane>>5)*8 + j]
// B[16][32]: lane holds B[(lane>>5)*8 + j][lane&31]
// C[32][32]: lane holds C[(reg&3) + 8*(reg>>2) + 4*(lane>>5)][lane&31]
typedef __attribute__((ext_vector_type(16))) float f32x16;

__global__ void mfma_probe32_kernel(const bf16* __restrict__ A,
                                    const bf16* __restrict__ B,
                                    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5, lo = lane & 31;
  const short8 a = *reinterpret_cast<const short8*>(A + lo * 16 + hi * 8);
  short8 b;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    reinterpret_cast<short*>(&b)[j] =
        reinterpret_cast<const short*>(B)[(hi * 8 + j) * 32 + lo];
  f32x16 c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    C[((r & 3) + 8 * (r >> 2) + 4 * hi) * 32 + lo] = c[r];
}

PRIME_API int prime_mfma_probe32(hipStream_t stream, const void* A,
                                 const void* B, void* C) {
  hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, stream,
                     (const bf16*)A, (const bf16*)B, (float*)C);
  return (int)hipGetLastError();
}
