// Single-token decode attention for gfx950 (serving path).
//
// Memory-bound: the cost is streaming the KV cache once per step. One
// 128-thread block per (batch, query-head); flash-style chunked online
// softmax; V rows are read COALESCED (thread t owns output dim d=t, all
// threads read V[l][0..D) together); K rows are read per-thread as
// bf16x8 chunks against q staged in LDS.
//
// Q [B,H,D], K/V cache [B,Smax,Hkv,D] (row stride Hkv*D), O [B,H,D],
// attend the first L rows. D in {64,128}.
#include "common.h"

#define DEC_CHUNK 256

template <int D>
__global__ __launch_bounds__(128) void attn_decode_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O, int B, int H, int Hkv,
    int Smax, int L, float scale, const int* __restrict__ len_dev) {
  if (len_dev) L = *len_dev;  // hipGraph decode: dynamic cache length
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh - b * H;
  const int hkv = h / (H / Hkv);
  const int t = threadIdx.x;
  const int64_t row_stride = (int64_t)Hkv * D;

  __shared__ float q_lds[D];
  __shared__ float p_lds[DEC_CHUNK];
  __shared__ float red[16];

  const bf16* qp = Q + ((int64_t)b * H + h) * D;
  const bf16* Kb = K + (int64_t)b * Smax * row_stride + (int64_t)hkv * D;
  const bf16* Vb = V + (int64_t)b * Smax * row_stride + (int64_t)hkv * D;

  for (int i = t; i < D; i += 128) q_lds[i] = bf2f(qp[i]);
  __syncthreads();

  float m = -1e30f, lsum = 0.f;
  float acc = 0.f;  // output accumulator for my dim d = t (t < D)
  const int d = t & (D - 1);
  const int dup = 128 / D;  // threads per dim (1 for D=128, 2 for D=64)

  for (int c0 = 0; c0 < L; c0 += DEC_CHUNK) {
    const int cn = min(DEC_CHUNK, L - c0);
    // ---- scores for this chunk (2 keys per thread)
    float smax = -1e30f;
    float sv[DEC_CHUNK / 128];
#pragma unroll
    for (int i = 0; i < DEC_CHUNK / 128; ++i) {
      const int l = i * 128 + t;
      float s = -1e30f;
      if (l < cn) {
        const bf16x8* kr = reinterpret_cast<const bf16x8*>(Kb + (int64_t)(c0 + l) * row_stride);
        float dot = 0.f;
#pragma unroll
        for (int j = 0; j < D / 8; ++j) {
          const bf16x8 kv8 = kr[j];
#pragma unroll
          for (int u = 0; u < 8; ++u) dot += q_lds[j * 8 + u] * bf2f(kv8.v[u]);
        }
        s = dot * scale;
      }
      sv[i] = s;
      smax = fmaxf(smax, s);
    }
    smax = block_reduce_max(smax, red);
    const float mnew = fmaxf(m, smax);
    const float alpha = __expf(m - mnew);
    float psum = 0.f;
#pragma unroll
    for (int i = 0; i < DEC_CHUNK / 128; ++i) {
      const int l = i * 128 + t;
      const float p = (l < cn) ? __expf(sv[i] - mnew) : 0.f;
      if (l < DEC_CHUNK) p_lds[l] = p;
      psum += p;
    }
    psum = block_reduce_sum(psum, red);
    lsum = lsum * alpha + psum;
    m = mnew;
    acc *= alpha;
    __syncthreads();  // p_lds ready
    // ---- accumulate V rows (coalesced: threads cover dims; duplicate
    // thread groups split the l range and are combined at the end)
    for (int l = t / D; l < cn; l += dup) {
      const float p = p_lds[l];
      if (p != 0.f)
        acc += p * bf2f(Vb[(int64_t)(c0 + l) * row_stride + d]);
    }
    __syncthreads();  // p_lds reuse next chunk
  }
  // combine duplicate threads per dim (D=64: two partial sums per d)
  float total = acc;
  if (dup == 2) {
    __shared__ float part[128];
    part[t] = acc;
    __syncthreads();
    if (t < D) total = part[t] + part[t + D];
  }
  if (t < D)
    O[((int64_t)b * H + h) * D + d] = f2bf(total / fmaxf(lsum, 1e-30f));
}

PRIME_API int prime_attn_decode(hipStream_t stream, const void* Q,
                                const void* K, const void* V, void* O,
                                int64_t B, int64_t H, int64_t Hkv,
                                int64_t Smax, int64_t L, int64_t D,
                                double scale, const void* len_dev) {
  if ((D != 64 && D != 128) || L < 1) return hipErrorInvalidValue;
  const int grid = (int)(B * H);
  if (D == 128)
    hipLaunchKernelGGL(attn_decode_kernel<128>, dim3(grid), dim3(128), 0,
                       stream, (const bf16*)Q, (const bf16*)K, (const bf16*)V,
                       (bf16*)O, (int)B, (int)H, (int)Hkv, (int)Smax, (int)L,
                       (float)scale, (const int*)len_dev);
  else
    hipLaunchKernelGGL(attn_decode_kernel<64>, dim3(grid), dim3(128), 0,
                       stream, (const bf16*)Q, (const bf16*)K, (const bf16*)V,
                       (bf16*)O, (int)B, (int)H, (int)Hkv, (int)Smax, (int)L,
                       (float)scale, (const int*)len_dev);
  return (int)hipGetLastError();
}
