// Flash attention (causal/non-causal, GQA) for gfx950 — MFMA
// v_mfma_f32_16x16x32_bf16 tiles, online softmax, FA2-style split backward
// (dQ kernel with q-outer loop; dK/dV kernel with kv-outer loop; no
// atomics — each accumulator lives in registers).
//
// Layouts (chosen so EVERY global operand-fragment load is one contiguous
// 16-byte bf16x8 per lane — see fragment maps below):
//   fwd:  Q[B,H,S,D], K[B,Hkv,S,D], Vt[B,Hkv,D,S]  ->  O[B,H,S,D], lse[B,H,S]
//   bwd:  additionally Kt[B,Hkv,D,S], Qt[B,H,D,S], dO[B,H,S,D], dOt[B,H,D,S],
//         delta[B,H,S] (rowsum(dO*O), precomputed)
// S must be a multiple of 64 (checked host-side).
//
// Fragment maps for mfma_f32_16x16x32_bf16 (verified on HW by
// tests/test_ops_gpu.py::test_mfma_layout against torch.matmul):
//   A[16][32]: lane l holds A[l%16][(l/16)*8 + j]          j = 0..7
//   B[32][16]: lane l holds B[(l/16)*8 + j][l%16]
//   C[16][16]: lane l holds C[(l/16)*4 + r][l%16]          r = 0..3
//
// Workgroup = 4 waves; each wave owns 16 q-rows (fwd/dQ) or 16 k-rows
// (dK/dV); tile = 64 x 64. P/dS cross-layout hops (C-layout -> A-layout)
// bounce through a per-wave LDS [16][64] bf16 buffer.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ short8 ld8(const bf16* p) {
  return *reinterpret_cast<const short8*>(p);
}

__device__ __forceinline__ f32x4 mfma16(short8 a, short8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// reduce over the 16-lane group (low 4 bits of lane id)
__device__ __forceinline__ float grp16_max(float x) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}
__device__ __forceinline__ float grp16_sum(float x) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) x += __shfl_xor(x, off, 64);
  return x;
}

#define NEG_INF (-1e30f)

// ---------------------------------------------------------------- forward
template <int D>
__global__ __launch_bounds__(256) void flash_fwd_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ Vt, bf16* __restrict__ O, float* __restrict__ lse,
    int B, int H, int Hkv, int S, float scale, int causal) {
  constexpr int DS = D / 32;   // 32-wide d slices (MFMA K dim)
  constexpr int DT = D / 16;   // 16-wide d subtiles (output cols)
  const int n_qt = S / 64;
  const int bh = blockIdx.x / n_qt;
  const int qt = blockIdx.x - bh * n_qt;
  const int b = bh / H, h = bh - b * H;
  const int hkv = h / (H / Hkv);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;   // 4-group index (0..3)
  const int li = lane & 15;   // index within 16-lane group

  const int q0 = qt * 64 + wid * 16;  // this wave's first q row
  const bf16* Qb = Q + ((int64_t)(b * H + h) * S) * D;
  const bf16* Kb = K + ((int64_t)(b * Hkv + hkv) * S) * D;
  const bf16* Vtb = Vt + ((int64_t)(b * Hkv + hkv) * D) * S;

  __shared__ bf16 p_lds_all[4][16][64];
  bf16(*p_lds)[64] = p_lds_all[wid];

  // Q fragments, resident for the whole block
  short8 qf[DS];
#pragma unroll
  for (int ds = 0; ds < DS; ++ds)
    qf[ds] = ld8(Qb + (int64_t)(q0 + li) * D + ds * 32 + lg * 8);

  float m[4], l[4];
  f32x4 o_acc[DT];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m[r] = NEG_INF; l[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) o_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? (qt * 64 + 64) : S;
  for (int kv = 0; kv < kv_end; kv += 64) {
    // ---- S = scale * Q K^T over four 16-key subtiles
    f32x4 s[4];
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 acc{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ds = 0; ds < DS; ++ds) {
        const short8 kf = ld8(Kb + (int64_t)(kv + sub * 16 + li) * D + ds * 32 + lg * 8);
        acc = mfma16(qf[ds], kf, acc);
      }
      s[sub] = acc;
    }
    // ---- online softmax (per q-row = per reg r; row owned by 16-lane group)
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qg = q0 + lg * 4 + r;
      float tmax = NEG_INF;
#pragma unroll
      for (int sub = 0; sub < 4; ++sub) {
        float v = s[sub][r] * scale;
        if (causal && (kv + sub * 16 + li) > qg) v = NEG_INF;
        s[sub][r] = v;
        tmax = fmaxf(tmax, v);
      }
      tmax = grp16_max(tmax);
      const float mnew = fmaxf(m[r], tmax);
      alpha[r] = __expf(m[r] - mnew);
      float psum = 0.f;
#pragma unroll
      for (int sub = 0; sub < 4; ++sub) {
        const float p = __expf(s[sub][r] - mnew);
        s[sub][r] = p;
        psum += p;
      }
      l[r] = l[r] * alpha[r] + grp16_sum(psum);
      m[r] = mnew;
    }
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[dt][r] *= alpha[r];
    // ---- P (C-layout) -> LDS (A-layout source)
    __syncthreads();  // reads of previous iteration's P are done
#pragma unroll
    for (int sub = 0; sub < 4; ++sub)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[lg * 4 + r][sub * 16 + li] = f2bf(s[sub][r]);
    __syncthreads();
    // ---- O += P V  (A from LDS, B from Vt: both contiguous 16B)
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const short8 pa = ld8(&p_lds[li][ks * 32 + lg * 8]);
        const short8 vb = ld8(Vtb + (int64_t)(dt * 16 + li) * S + kv + ks * 32 + lg * 8);
        o_acc[dt] = mfma16(pa, vb, o_acc[dt]);
      }
    }
  }
  // ---- epilogue: O /= l ; lse = m + log(l)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv = (l[r] > 0.f) ? 1.f / l[r] : 0.f;
    const int qg = q0 + lg * 4 + r;
    bf16* orow = O + (((int64_t)(b * H + h) * S) + qg) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
      orow[dt * 16 + li] = f2bf(o_acc[dt][r] * inv);
    if (li == 0)
      lse[((int64_t)(b * H + h) * S) + qg] = m[r] + __logf(fmaxf(l[r], 1e-30f));
  }
}

// ------------------------------------------------------- delta = rowsum(dO*O)
__global__ void attn_delta_kernel(const bf16* __restrict__ dO,
                                  const bf16* __restrict__ O,
                                  float* __restrict__ delta, int64_t rows,
                                  int D) {
  // one wave per row
  const int64_t row0 = (int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  for (int64_t r = row0; r < rows; r += (int64_t)gridDim.x * (blockDim.x >> 6)) {
    float acc = 0.f;
    for (int i = lane * 2; i < D; i += 128) {
      acc += bf2f(dO[r * D + i]) * bf2f(O[r * D + i]);
      acc += bf2f(dO[r * D + i + 1]) * bf2f(O[r * D + i + 1]);
    }
    acc = wave_reduce_sum(acc);
    acc = __shfl(acc, 0, 64);
    if (lane == 0) delta[r] = acc;
  }
}

// ----------------------------------------------------------- backward dQ
template <int D>
__global__ __launch_bounds__(256) void flash_bwd_dq_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ Kt, const bf16* __restrict__ V,
    const bf16* __restrict__ dO, const float* __restrict__ lse,
    const float* __restrict__ delta, bf16* __restrict__ dQ, int B, int H,
    int Hkv, int S, float scale, int causal) {
  constexpr int DS = D / 32;
  constexpr int DT = D / 16;
  const int n_qt = S / 64;
  const int bh = blockIdx.x / n_qt;
  const int qt = blockIdx.x - bh * n_qt;
  const int b = bh / H, h = bh - b * H;
  const int hkv = h / (H / Hkv);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  const int q0 = qt * 64 + wid * 16;

  const bf16* Qb = Q + ((int64_t)(b * H + h) * S) * D;
  const bf16* dOb = dO + ((int64_t)(b * H + h) * S) * D;
  const bf16* Kb = K + ((int64_t)(b * Hkv + hkv) * S) * D;
  const bf16* Ktb = Kt + ((int64_t)(b * Hkv + hkv) * D) * S;
  const bf16* Vb = V + ((int64_t)(b * Hkv + hkv) * S) * D;
  const float* lse_b = lse + (int64_t)(b * H + h) * S;
  const float* dl_b = delta + (int64_t)(b * H + h) * S;

  __shared__ bf16 ds_lds_all[4][16][64];
  bf16(*ds_lds)[64] = ds_lds_all[wid];

  short8 qf[DS], dof[DS];
#pragma unroll
  for (int ds = 0; ds < DS; ++ds) {
    qf[ds] = ld8(Qb + (int64_t)(q0 + li) * D + ds * 32 + lg * 8);
    dof[ds] = ld8(dOb + (int64_t)(q0 + li) * D + ds * 32 + lg * 8);
  }
  float lse_r[4], dl_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    lse_r[r] = lse_b[q0 + lg * 4 + r];
    dl_r[r] = dl_b[q0 + lg * 4 + r];
  }
  f32x4 dq_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) dq_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? (qt * 64 + 64) : S;
  for (int kv = 0; kv < kv_end; kv += 64) {
    __syncthreads();  // previous iteration's ds_lds reads done
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 s_acc{0.f, 0.f, 0.f, 0.f}, dp_acc{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ds = 0; ds < DS; ++ds) {
        const bf16* krow = Kb + (int64_t)(kv + sub * 16 + li) * D + ds * 32 + lg * 8;
        const bf16* vrow = Vb + (int64_t)(kv + sub * 16 + li) * D + ds * 32 + lg * 8;
        s_acc = mfma16(qf[ds], ld8(krow), s_acc);
        dp_acc = mfma16(dof[ds], ld8(vrow), dp_acc);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qg = q0 + lg * 4 + r;
        const int kg = kv + sub * 16 + li;
        float p = __expf(s_acc[r] * scale - lse_r[r]);
        if (causal && kg > qg) p = 0.f;
        const float dsv = p * (dp_acc[r] - dl_r[r]) * scale;
        ds_lds[lg * 4 + r][sub * 16 + li] = f2bf(dsv);
      }
    }
    __syncthreads();
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const short8 a = ld8(&ds_lds[li][ks * 32 + lg * 8]);
        const short8 bfr = ld8(Ktb + (int64_t)(dt * 16 + li) * S + kv + ks * 32 + lg * 8);
        dq_acc[dt] = mfma16(a, bfr, dq_acc[dt]);
      }
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    bf16* row = dQ + (((int64_t)(b * H + h) * S) + q0 + lg * 4 + r) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) row[dt * 16 + li] = f2bf(dq_acc[dt][r]);
  }
}

// -------------------------------------------------------- backward dK, dV
template <int D>
__global__ __launch_bounds__(256) void flash_bwd_dkv_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ Qt,
    const bf16* __restrict__ K, const bf16* __restrict__ V,
    const bf16* __restrict__ dO, const bf16* __restrict__ dOt,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dK, bf16* __restrict__ dV, int B, int H, int Hkv,
    int S, float scale, int causal) {
  constexpr int DS = D / 32;
  constexpr int DT = D / 16;
  const int n_kt = S / 64;
  const int bh = blockIdx.x / n_kt;
  const int kt = blockIdx.x - bh * n_kt;
  const int b = bh / Hkv, hkv = bh - b * Hkv;
  const int group = H / Hkv;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  const int k0 = kt * 64 + wid * 16;  // this wave's first key row

  const bf16* Kb = K + ((int64_t)(b * Hkv + hkv) * S) * D;
  const bf16* Vb = V + ((int64_t)(b * Hkv + hkv) * S) * D;

  __shared__ bf16 pt_lds_all[4][16][64];
  __shared__ bf16 dst_lds_all[4][16][64];
  bf16(*pt_lds)[64] = pt_lds_all[wid];
  bf16(*dst_lds)[64] = dst_lds_all[wid];

  short8 kf[DS], vf[DS];
#pragma unroll
  for (int ds = 0; ds < DS; ++ds) {
    kf[ds] = ld8(Kb + (int64_t)(k0 + li) * D + ds * 32 + lg * 8);
    vf[ds] = ld8(Vb + (int64_t)(k0 + li) * D + ds * 32 + lg * 8);
  }
  f32x4 dk_acc[DT], dv_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) {
    dk_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  for (int g = 0; g < group; ++g) {
    const int h = hkv * group + g;
    const bf16* Qb = Q + ((int64_t)(b * H + h) * S) * D;
    const bf16* Qtb = Qt + ((int64_t)(b * H + h) * D) * S;
    const bf16* dOb = dO + ((int64_t)(b * H + h) * S) * D;
    const bf16* dOtb = dOt + ((int64_t)(b * H + h) * D) * S;
    const float* lse_b = lse + (int64_t)(b * H + h) * S;
    const float* dl_b = delta + (int64_t)(b * H + h) * S;
    const int q_start = causal ? kt * 64 : 0;
    for (int q0g = q_start; q0g < S; q0g += 64) {
      __syncthreads();
#pragma unroll
      for (int sub = 0; sub < 4; ++sub) {
        f32x4 st_acc{0.f, 0.f, 0.f, 0.f}, dpt_acc{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ds = 0; ds < DS; ++ds) {
          // B operands: column q = li, rows d -> contiguous in Q / dO rows
          const short8 qb = ld8(Qb + (int64_t)(q0g + sub * 16 + li) * D + ds * 32 + lg * 8);
          const short8 dob = ld8(dOb + (int64_t)(q0g + sub * 16 + li) * D + ds * 32 + lg * 8);
          st_acc = mfma16(kf[ds], qb, st_acc);
          dpt_acc = mfma16(vf[ds], dob, dpt_acc);
        }
        const int qg = q0g + sub * 16 + li;
        const float lse_q = lse_b[qg];
        const float dl_q = dl_b[qg];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kg = k0 + lg * 4 + r;
          float p = __expf(st_acc[r] * scale - lse_q);
          if (causal && kg > qg) p = 0.f;
          pt_lds[lg * 4 + r][sub * 16 + li] = f2bf(p);
          dst_lds[lg * 4 + r][sub * 16 + li] = f2bf(p * (dpt_acc[r] - dl_q) * scale);
        }
      }
      __syncthreads();
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const short8 pa = ld8(&pt_lds[li][ks * 32 + lg * 8]);
          const short8 da = ld8(&dst_lds[li][ks * 32 + lg * 8]);
          const short8 dob = ld8(dOtb + (int64_t)(dt * 16 + li) * S + q0g + ks * 32 + lg * 8);
          const short8 qtb = ld8(Qtb + (int64_t)(dt * 16 + li) * S + q0g + ks * 32 + lg * 8);
          dv_acc[dt] = mfma16(pa, dob, dv_acc[dt]);
          dk_acc[dt] = mfma16(da, qtb, dk_acc[dt]);
        }
      }
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kg = k0 + lg * 4 + r;
    bf16* krow = dK + (((int64_t)(b * Hkv + hkv) * S) + kg) * D;
    bf16* vrow = dV + (((int64_t)(b * Hkv + hkv) * S) + kg) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      krow[dt * 16 + li] = f2bf(dk_acc[dt][r]);
      vrow[dt * 16 + li] = f2bf(dv_acc[dt][r]);
    }
  }
}

// ------------------------------------------------------------------ host API
PRIME_API int prime_flash_fwd(hipStream_t stream, const void* Q, const void* K,
                              const void* Vt, void* O, void* lse, int64_t B,
                              int64_t H, int64_t Hkv, int64_t S, int64_t D,
                              double scale, int64_t causal) {
  if (S % 64 != 0 || (D != 64 && D != 128)) return hipErrorInvalidValue;
  const int grid = (int)(B * H * (S / 64));
  if (D == 128)
    hipLaunchKernelGGL(flash_fwd_kernel<128>, dim3(grid), dim3(256), 0, stream,
                       (const bf16*)Q, (const bf16*)K, (const bf16*)Vt,
                       (bf16*)O, (float*)lse, (int)B, (int)H, (int)Hkv, (int)S,
                       (float)scale, (int)causal);
  else
    hipLaunchKernelGGL(flash_fwd_kernel<64>, dim3(grid), dim3(256), 0, stream,
                       (const bf16*)Q, (const bf16*)K, (const bf16*)Vt,
                       (bf16*)O, (float*)lse, (int)B, (int)H, (int)Hkv, (int)S,
                       (float)scale, (int)causal);
  return (int)hipGetLastError();
}

PRIME_API int prime_attn_delta(hipStream_t stream, const void* dO,
                               const void* O, void* delta, int64_t rows,
                               int64_t D) {
  int grid = prime_grid(rows * 64, 256);
  hipLaunchKernelGGL(attn_delta_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bf16*)dO, (const bf16*)O, (float*)delta, rows,
                     (int)D);
  return (int)hipGetLastError();
}

PRIME_API int prime_flash_bwd_dq(hipStream_t stream, const void* Q,
                                 const void* K, const void* Kt, const void* V,
                                 const void* dO, const void* lse,
                                 const void* delta, void* dQ, int64_t B,
                                 int64_t H, int64_t Hkv, int64_t S, int64_t D,
                                 double scale, int64_t causal) {
  if (S % 64 != 0 || (D != 64 && D != 128)) return hipErrorInvalidValue;
  const int grid = (int)(B * H * (S / 64));
  if (D == 128)
    hipLaunchKernelGGL(flash_bwd_dq_kernel<128>, dim3(grid), dim3(256), 0,
                       stream, (const bf16*)Q, (const bf16*)K, (const bf16*)Kt,
                       (const bf16*)V, (const bf16*)dO, (const float*)lse,
                       (const float*)delta, (bf16*)dQ, (int)B, (int)H,
                       (int)Hkv, (int)S, (float)scale, (int)causal);
  else
    hipLaunchKernelGGL(flash_bwd_dq_kernel<64>, dim3(grid), dim3(256), 0,
                       stream, (const bf16*)Q, (const bf16*)K, (const bf16*)Kt,
                       (const bf16*)V, (const bf16*)dO, (const float*)lse,
                       (const float*)delta, (bf16*)dQ, (int)B, (int)H,
                       (int)Hkv, (int)S, (float)scale, (int)causal);
  return (int)hipGetLastError();
}

PRIME_API int prime_flash_bwd_dkv(hipStream_t stream, const void* Q,
                                  const void* Qt, const void* K, const void* V,
                                  const void* dO, const void* dOt,
                                  const void* lse, const void* delta, void* dK,
                                  void* dV, int64_t B, int64_t H, int64_t Hkv,
                                  int64_t S, int64_t D, double scale,
                                  int64_t causal) {
  if (S % 64 != 0 || (D != 64 && D != 128)) return hipErrorInvalidValue;
  const int grid = (int)(B * Hkv * (S / 64));
  if (D == 128)
    hipLaunchKernelGGL(flash_bwd_dkv_kernel<128>, dim3(grid), dim3(256), 0,
                       stream, (const bf16*)Q, (const bf16*)Qt, (const bf16*)K,
                       (const bf16*)V, (const bf16*)dO, (const bf16*)dOt,
                       (const float*)lse, (const float*)delta, (bf16*)dK,
                       (bf16*)dV, (int)B, (int)H, (int)Hkv, (int)S,
                       (float)scale, (int)causal);
  else
    hipLaunchKernelGGL(flash_bwd_dkv_kernel<64>, dim3(grid), dim3(256), 0,
                       stream, (const bf16*)Q, (const bf16*)Qt, (const bf16*)K,
                       (const bf16*)V, (const bf16*)dO, (const bf16*)dOt,
                       (const float*)lse, (const float*)delta, (bf16*)dK,
                       (bf16*)dV, (int)B, (int)H, (int)Hkv, (int)S,
                       (float)scale, (int)causal);
  return (int)hipGetLastError();
}
