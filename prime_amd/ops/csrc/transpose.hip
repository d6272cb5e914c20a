// Batched bf16 transpose [B,S,H,D] (strided) -> [B,H,D,S] (contiguous)
// for the attention operands that genuinely need a transposed copy
// (Vt/Kt/Qt/dOt: contiguous MFMA B-fragments). PyTorch's
// permute().contiguous() runs this pattern at ~80 GB/s on ROCm; this
// LDS-tiled version (64x64 tiles, 16 B loads AND stores, padded LDS rows)
// is HBM-bound (~5 TB/s class).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;

#define TP 72  // padded LDS row: 64 + 8 elems (16 B) keeps both phases conflict-light

__global__ __launch_bounds__(256) void transpose_bshd_kernel(
    const bf16* __restrict__ src, bf16* __restrict__ dst, int B, int S, int H,
    int D, int64_t sb, int64_t ss, int64_t sh) {
  __shared__ bf16 tile[64][TP];
  const int n_st = S / 64, n_dt = D / 64;
  int idx = blockIdx.x;
  const int dt = idx % n_dt; idx /= n_dt;
  const int st = idx % n_st; idx /= n_st;
  const int h = idx % H; idx /= H;
  const int b = idx;
  const bf16* sbase = src + b * sb + h * sh + (int64_t)(st * 64) * ss + dt * 64;
  bf16* dbase = dst + (((int64_t)(b * H + h) * D) + dt * 64) * S + st * 64;

  const int t = threadIdx.x;
  // load 64 rows x 64 cols: 512 x 8-elem units, 2 per thread
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int u = i * 256 + t;
    const int r = u >> 3, c8 = (u & 7) * 8;
    *reinterpret_cast<short8*>(&tile[r][c8]) =
        *reinterpret_cast<const short8*>(sbase + (int64_t)r * ss + c8);
  }
  __syncthreads();
  // store transposed: dst row = d (64), cols = s (64)
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int u = i * 256 + t;
    const int d = u >> 3, s8 = (u & 7) * 8;
    short8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      reinterpret_cast<short*>(&v)[j] =
          reinterpret_cast<const short*>(&tile[s8 + j][d])[0];
    *reinterpret_cast<short8*>(dbase + (int64_t)d * S + s8) = v;
  }
}

PRIME_API int prime_transpose_bshd(hipStream_t stream, const void* src,
                                   void* dst, int64_t B, int64_t S, int64_t H,
                                   int64_t D, int64_t sb, int64_t ss,
                                   int64_t sh) {
  if (S % 64 != 0 || D % 64 != 0) return hipErrorInvalidValue;
  const int64_t grid = B * H * (S / 64) * (D / 64);
  if (grid > 0x7fffffff) return hipErrorInvalidValue;
  hipLaunchKernelGGL(transpose_bshd_kernel, dim3((int)grid), dim3(256), 0,
                     stream, (const bf16*)src, (bf16*)dst, (int)B, (int)S,
                     (int)H, (int)D, sb, ss, sh);
  return (int)hipGetLastError();
}

// transpose + fp8 quantize in one pass (wgrad operands x^T / dY^T are
// consumed ONLY as fp8 in the fp8-wgrad tier — the bf16 intermediate
// was pure traffic). Same 64x64 LDS tiling; scale comes from the
// delayed-scaling amax state (device scalar), this tile's amax
// accumulates for the next step.
#include <hip/hip_fp8.h>

template <int E5M2>
__global__ __launch_bounds__(256) void transpose_fp8_kernel(
    const bf16* __restrict__ src, unsigned char* __restrict__ dst, int B,
    int S, int H, int D, int64_t sb, int64_t ss, int64_t sh,
    const float* __restrict__ amax_prev, float* __restrict__ amax_next,
    float* __restrict__ sinv_out) {
  __shared__ bf16 tile[64][TP];
  const float fmax8 = E5M2 ? 57344.f : 448.f;
  const float prev = fmaxf(*amax_prev, 1e-12f);
  const float s = fmax8 / prev;
  if (blockIdx.x == 0 && threadIdx.x == 0) sinv_out[0] = prev / fmax8;
  const int n_st = S / 64, n_dt = D / 64;
  int idx = blockIdx.x;
  const int dt = idx % n_dt; idx /= n_dt;
  const int st = idx % n_st; idx /= n_st;
  const int h = idx % H; idx /= H;
  const int b = idx;
  const bf16* sbase = src + b * sb + h * sh + (int64_t)(st * 64) * ss + dt * 64;
  unsigned char* dbase =
      dst + (((int64_t)(b * H + h) * D) + dt * 64) * S + st * 64;

  const int t = threadIdx.x;
  float amax = 0.f;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int u = i * 256 + t;
    const int r = u >> 3, c8 = (u & 7) * 8;
    *reinterpret_cast<short8*>(&tile[r][c8]) =
        *reinterpret_cast<const short8*>(sbase + (int64_t)r * ss + c8);
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int u = i * 256 + t;
    const int d = u >> 3, s8 = (u & 7) * 8;
    unsigned long long packed = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float v = bf2f(tile[s8 + j][d]);
      amax = fmaxf(amax, fabsf(v));
      const float q = fminf(fmaxf(v * s, -fmax8), fmax8);
      unsigned char bq = E5M2 ? __hip_fp8_e5m2(q).__x : __hip_fp8_e4m3(q).__x;
      packed |= (unsigned long long)bq << (8 * j);
    }
    *reinterpret_cast<unsigned long long*>(dbase + (int64_t)d * S + s8) = packed;
  }
  amax = wave_reduce_max(amax);
  __shared__ float wmax[4];
  const int wid = t >> 6;
  if ((t & 63) == 0) wmax[wid] = amax;
  __syncthreads();
  if (t == 0) {
    float m = wmax[0];
#pragma unroll
    for (int i = 1; i < 4; ++i) m = fmaxf(m, wmax[i]);
    atomicMax(reinterpret_cast<unsigned*>(amax_next), __float_as_uint(m));
  }
}

PRIME_API int prime_transpose_fp8(hipStream_t stream, const void* src,
                                  void* dst, int64_t B, int64_t S, int64_t H,
                                  int64_t D, int64_t sb, int64_t ss,
                                  int64_t sh, const void* amax_prev,
                                  void* amax_next, void* sinv_out,
                                  int64_t e5m2) {
  if (S % 64 != 0 || D % 64 != 0) return hipErrorInvalidValue;
  const int64_t grid = B * H * (S / 64) * (D / 64);
  if (grid > 0x7fffffff) return hipErrorInvalidValue;
  if (e5m2)
    hipLaunchKernelGGL((transpose_fp8_kernel<1>), dim3((int)grid), dim3(256),
                       0, stream, (const bf16*)src, (unsigned char*)dst,
                       (int)B, (int)S, (int)H, (int)D, sb, ss, sh,
                       (const float*)amax_prev, (float*)amax_next,
                       (float*)sinv_out);
  else
    hipLaunchKernelGGL((transpose_fp8_kernel<0>), dim3((int)grid), dim3(256),
                       0, stream, (const bf16*)src, (unsigned char*)dst,
                       (int)B, (int)S, (int)H, (int)D, sb, ss, sh,
                       (const float*)amax_prev, (float*)amax_next,
                       (float*)sinv_out);
  return (int)hipGetLastError();
}
