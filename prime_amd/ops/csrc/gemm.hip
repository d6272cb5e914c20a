// bf16 GEMM for gfx950 — C[M,N] = A[M,K] · B[N,K]^T (the "NT" layout of
// every torch F.linear forward), fp32 accumulation, hand-written MFMA.
//
// Structure (guide §5 "256² 8-phase template", re-derived for this code
// base rather than copied: 256x256 output tile, BK=64, 8 waves (2M x 4N,
// 128x64 per wave), double-buffered XOR-16B-swizzled LDS staged with
// width-16 global_load_lds, prefetch-behind-barrier so each tile's loads
// get the whole previous tile's 64 MFMAs to land, ONE __syncthreads per
// K-tile (vmcnt drain + rendezvous in one place — the m97-style
// stage;sync;compute order drains with zero issue-to-wait gap, which is
// the ~20% stall the guide documents).
//
// Dispatch policy lives host-side in functional.py: this kernel requires
// M%256==0, N%256==0, K%64==0 and beats hipBLASLt on the large training
// shapes (measured per shape by tools/perf_gemm.py); anything else falls
// back to the library.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((address_space(3))) char g_lds_char;
typedef __attribute__((address_space(3))) void g_lds_void;
typedef __attribute__((address_space(1))) const void g_g_void;

__device__ __forceinline__ f32x4 g_mfma16(short8 a, short8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// stage a [256][64] bf16 tile (rows from a strided global matrix) into
// linear LDS with the T2 XOR-16B swizzle pre-applied to the SOURCE
// address (global_load_lds writes linearly: wave-uniform base + lane*16)
template <int NT>
__device__ __forceinline__ void g_stage_256x64(
    const bf16* __restrict__ gbase, int64_t row_stride, bf16* lds_tile,
    int tid) {
  constexpr int UNITS = 256 * 64 / 8;  // 16 B units
  constexpr int UPR = 8;               // units per row
#pragma unroll
  for (int i = 0; i < UNITS / NT; ++i) {
    const int u = i * NT + tid;
    const int row = u / UPR;
    const int colb = ((u % UPR) * 16) ^ ((row & 7) << 4);
    const bf16* src = gbase + (int64_t)row * row_stride + colb / 2;
    const int wid = tid >> 6;
    g_lds_void* dst =
        (g_lds_void*)((g_lds_char*)lds_tile + i * (NT * 16) + wid * 1024);
    __builtin_amdgcn_global_load_lds((g_g_void*)src, dst, 16, 0, 0);
  }
}

__device__ __forceinline__ short8 g_ld8_swz(const bf16* lds_tile, int row,
                                            int colb) {
  return *reinterpret_cast<const short8*>(
      reinterpret_cast<const char*>(lds_tile) + row * 128 +
      (colb ^ ((row & 7) << 4)));
}

template <int SETPRIO>
__global__ __launch_bounds__(512, 2) void gemm_nt_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    bf16* __restrict__ C, int M, int N, int K) {
  const int tiles_n = N / 256;
  // XCD-aware bijective remap: consecutive virtual blocks walk N-tiles
  // fastest (sharing the A panel), and each XCD gets a contiguous chunk
  const int nwg = gridDim.x;
  int vb = blockIdx.x;
  if ((nwg & 7) == 0) vb = (blockIdx.x & 7) * (nwg >> 3) + (blockIdx.x >> 3);
  const int tm = vb / tiles_n;
  const int tn = vb - tm * tiles_n;

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  const int wm = wid >> 2;       // 0..1: M half
  const int wn = wid & 3;        // 0..3: N quarter
  const int am0 = wm * 128;      // wave's A rows within the tile
  const int bn0 = wn * 64;       // wave's B rows (= C cols) within the tile

  __shared__ bf16 a_lds[2][256 * 64];
  __shared__ bf16 b_lds[2][256 * 64];

  const bf16* Ab = A + (int64_t)(tm * 256) * K;
  const bf16* Bb = B + (int64_t)(tn * 256) * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int mf = 0; mf < 8; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) acc[mf][nf] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nk = K / 64;
  g_stage_256x64<512>(Ab, K, a_lds[0], threadIdx.x);
  g_stage_256x64<512>(Bb, K, b_lds[0], threadIdx.x);
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    __syncthreads();  // prev buffer free + this tile's stage drained
    if (kt + 1 < nk) {
      g_stage_256x64<512>(Ab + (kt + 1) * 64, K, a_lds[cur ^ 1], threadIdx.x);
      g_stage_256x64<512>(Bb + (kt + 1) * 64, K, b_lds[cur ^ 1], threadIdx.x);
    }
    if (SETPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      short8 af[8];
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
        af[mf] = g_ld8_swz(a_lds[cur], am0 + mf * 16 + li, ks * 64 + lg * 16);
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const short8 bf =
            g_ld8_swz(b_lds[cur], bn0 + nf * 16 + li, ks * 64 + lg * 16);
#pragma unroll
        for (int mf = 0; mf < 8; ++mf)
          acc[mf][nf] = g_mfma16(af[mf], bf, acc[mf][nf]);
      }
    }
    if (SETPRIO) __builtin_amdgcn_s_setprio(0);
  }
  // ---- epilogue: C rows tm*256 + am0 + mf*16 + lg*4 + r
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t row = (int64_t)tm * 256 + am0 + mf * 16 + lg * 4 + r;
      bf16* crow = C + row * N + tn * 256 + bn0;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        crow[nf * 16 + li] = f2bf(acc[mf][nf][r]);
    }
  }
}

// ------------------------------------------------------------------ host
PRIME_API int prime_gemm_nt(hipStream_t stream, const void* A, const void* B,
                            void* C, int64_t M, int64_t N, int64_t K,
                            int64_t setprio) {
  if (M % 256 || N % 256 || K % 64) return hipErrorInvalidValue;
  const int grid = (int)((M / 256) * (N / 256));
  if (setprio)
    hipLaunchKernelGGL((gemm_nt_kernel<1>), dim3(grid), dim3(512), 0, stream,
                       (const bf16*)A, (const bf16*)B, (bf16*)C, (int)M,
                       (int)N, (int)K);
  else
    hipLaunchKernelGGL((gemm_nt_kernel<0>), dim3(grid), dim3(512), 0, stream,
                       (const bf16*)A, (const bf16*)B, (bf16*)C, (int)M,
                       (int)N, (int)K);
  return (int)hipGetLastError();
}

// ---------------------------------------------------------------- v2
// Counted-vmcnt phased schedule (guide §5.5 T3+T4+T5, re-derived with a
// provable half-open pipeline rather than a copy of the reference
// schedule): K advances in 32-wide sub-tiles with FOUR resident LDS
// slots; while sub-tile k computes, sub-tile k+3's stage is issued into
// the slot k-1 just vacated. The per-iteration boundary is
// `s_waitcnt vmcnt(8)` + raw s_barrier — the 8 newest loads (sub-tiles
// k+1, k+2) stay in flight across every barrier; the queue never drains
// to zero (the m97-structure stall this replaces). setprio(1) wraps the
// MFMA cluster (T5 pays once the schedule has wave role diversity).
__device__ __forceinline__ void g_stage_256x32(
    const bf16* __restrict__ gbase, int64_t row_stride, bf16* lds_tile,
    int tid) {
  constexpr int NT = 512;
  constexpr int UPR = 4;  // 16 B units per 64 B row
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int u = i * NT + tid;
    const int row = u / UPR;
    const int colb = ((u % UPR) * 16) ^ ((row & 3) << 4);
    const bf16* src = gbase + (int64_t)row * row_stride + colb / 2;
    const int wid = tid >> 6;
    g_lds_void* dst =
        (g_lds_void*)((g_lds_char*)lds_tile + i * (NT * 16) + wid * 1024);
    __builtin_amdgcn_global_load_lds((g_g_void*)src, dst, 16, 0, 0);
  }
}

__device__ __forceinline__ short8 g_ld8_swz32(const bf16* lds_tile, int row,
                                              int colb) {
  return *reinterpret_cast<const short8*>(
      reinterpret_cast<const char*>(lds_tile) + row * 64 +
      (colb ^ ((row & 3) << 4)));
}

template <int SETPRIO>
__global__ __launch_bounds__(512, 2) void gemm_nt8_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    bf16* __restrict__ C, int M, int N, int K) {
  const int tiles_n = N / 256;
  const int nwg = gridDim.x;
  int vb = blockIdx.x;
  if ((nwg & 7) == 0) vb = (blockIdx.x & 7) * (nwg >> 3) + (blockIdx.x >> 3);
  const int tm = vb / tiles_n;
  const int tn = vb - tm * tiles_n;

  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  const int wid = threadIdx.x >> 6;
  const int am0 = (wid >> 2) * 128;
  const int bn0 = (wid & 3) * 64;

  __shared__ bf16 a_lds[4][256 * 32];
  __shared__ bf16 b_lds[4][256 * 32];

  const bf16* Ab = A + (int64_t)(tm * 256) * K;
  const bf16* Bb = B + (int64_t)(tn * 256) * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int mf = 0; mf < 8; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) acc[mf][nf] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nk = K / 32;  // K sub-tiles
  // prologue: fill 3 slots (sub-tiles 0..2)
  for (int k = 0; k < 3 && k < nk; ++k) {
    g_stage_256x32(Ab + k * 32, K, a_lds[k], threadIdx.x);
    g_stage_256x32(Bb + k * 32, K, b_lds[k], threadIdx.x);
  }
  for (int k = 0; k < nk; ++k) {
    const int cur = k & 3;
    // boundary: sub-tile k's loads are 12 issues old; allow the newest 8
    // (k+1, k+2) to stay in flight
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
    if (k + 3 < nk) {
      g_stage_256x32(Ab + (k + 3) * 32, K, a_lds[(k + 3) & 3], threadIdx.x);
      g_stage_256x32(Bb + (k + 3) * 32, K, b_lds[(k + 3) & 3], threadIdx.x);
    }
    short8 af[8];
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
      af[mf] = g_ld8_swz32(a_lds[cur], am0 + mf * 16 + li, lg * 16);
    short8 bf[4];
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
      bf[nf] = g_ld8_swz32(b_lds[cur], bn0 + nf * 16 + li, lg * 16);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    if (SETPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
        acc[mf][nf] = g_mfma16(af[mf], bf[nf], acc[mf][nf]);
    if (SETPRIO) __builtin_amdgcn_s_setprio(0);
  }
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t row = (int64_t)tm * 256 + am0 + mf * 16 + lg * 4 + r;
      bf16* crow = C + row * N + tn * 256 + bn0;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        crow[nf * 16 + li] = f2bf(acc[mf][nf][r]);
    }
  }
}

PRIME_API int prime_gemm_nt8(hipStream_t stream, const void* A, const void* B,
                             void* C, int64_t M, int64_t N, int64_t K,
                             int64_t setprio) {
  if (M % 256 || N % 256 || K % 32) return hipErrorInvalidValue;
  const int grid = (int)((M / 256) * (N / 256));
  if (setprio)
    hipLaunchKernelGGL((gemm_nt8_kernel<1>), dim3(grid), dim3(512), 0, stream,
                       (const bf16*)A, (const bf16*)B, (bf16*)C, (int)M,
                       (int)N, (int)K);
  else
    hipLaunchKernelGGL((gemm_nt8_kernel<0>), dim3(grid), dim3(512), 0, stream,
                       (const bf16*)A, (const bf16*)B, (bf16*)C, (int)M,
                       (int)N, (int)K);
  return (int)hipGetLastError();
}
