// Flash attention (causal/non-causal, GQA) for gfx950 — MFMA
// v_mfma_f32_16x16x32_bf16 tiles, online softmax, FA2-style split backward
// (dQ q-outer; dK/dV kv-outer with a grid-split q loop + fp32 partial
// workspaces). A 32x32 swapped-operand forward (flash_fwd32_kernel) is
// kept behind PRIME_ATTN_V4=1 — measured occupancy-bound (see profiles).
//
// Design (profiled v1 at ~70 TF: redundant per-wave global reads and
// unswizzled LDS dominated; measured after fixes: fwd ~240 TF / bwd ~168
// TF at the 10B shape):
//  - K/V(+transposed) tiles are staged COOPERATIVELY once per block with
//    __builtin_amdgcn_global_load_lds width-16 (no VGPR round trip), with
//    the T2 XOR-16B swizzle applied via the pre-swizzled SOURCE address
//    (global_load_lds writes linearly: guide §5.5 T2 + rule 21).
//  - all operand-fragment LDS reads are ld8 (16 B) at the swizzled address:
//    2-way bank aliasing max (free on CDNA4).
//  - per-wave P / dS buffers are swizzled the same way (v1 left a 16-way
//    conflict on the PV A-fragment reads).
//  - kernels are STRIDE-AWARE over [B,S,H,D] inputs: q/k/v are consumed
//    directly as views into the packed qkv GEMM output, no transposes or
//    .contiguous() copies (v1 spent 8% of step time in copyBuffer).
//  - lse/delta layout [B,S,H] (contiguous with the row order of bshd).
//
// Fragment maps for mfma_f32_16x16x32_bf16 (HW-verified by
// tests/test_ops_gpu.py::test_mfma_layout_vs_matmul, asymmetric inputs):
//   A[16][32]: lane l holds A[l%16][(l/16)*8 + j]          j = 0..7
//   B[32][16]: lane l holds B[(l/16)*8 + j][l%16]
//   C[16][16]: lane l holds C[(l/16)*4 + r][l%16]          r = 0..3
//
// Workgroup = 4 waves; each wave owns 16 q-rows (fwd/dQ) or 16 k-rows
// (dK/dV); tile = 64 x 64. S % 64 == 0 (checked host-side), D in {64,128}.
#include "common.h"
#include <stdlib.h>

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((address_space(3))) char lds_char;
typedef __attribute__((address_space(3))) void lds_void;
typedef __attribute__((address_space(1))) const void g_void;

__device__ __forceinline__ short8 ld8(const bf16* p) {
  return *reinterpret_cast<const short8*>(p);
}

__device__ __forceinline__ f32x4 mfma16(short8 a, short8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

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

#ifndef PRIME_FWD_SETPRIO
#define PRIME_FWD_SETPRIO 0
#endif
#if PRIME_FWD_SETPRIO
#define FWD_PRIO(x) __builtin_amdgcn_s_setprio(x)
#else
#define FWD_PRIO(x)
#endif

// ---- swizzled-tile helpers --------------------------------------------
// A [ROWS][COLS] bf16 tile lives linearly in LDS; byte (row, colb) is
// stored at row*COLS*2 + (colb ^ ((row&7)<<4)). Stage from a strided
// global matrix; read 8-element fragments at the same swizzle.

template <int ROWS, int COLS, int NT = 256>
__device__ __forceinline__ void stage_tile(const bf16* __restrict__ gbase,
                                           int64_t row_stride, bf16* lds_tile,
                                           int tid) {
  constexpr int UNITS = ROWS * COLS / 8;  // 16 B units
  constexpr int UPR = COLS / 8;           // units per row
  static_assert(UNITS % NT == 0, "tile not divisible by block lanes");
#pragma unroll
  for (int i = 0; i < UNITS / NT; ++i) {
    const int u = i * NT + tid;
    const int row = u / UPR;
    const int colb = ((u % UPR) * 16) ^ ((row & 7) << 4);  // pre-swizzle src
    const bf16* src = gbase + (int64_t)row * row_stride + colb / 2;
    // linear dest: wave-uniform base + lane*16 (global_load_lds contract)
    const int wid = tid >> 6;
    lds_void* dst = (lds_void*)((lds_char*)lds_tile + i * (NT * 16) + wid * 1024);
    __builtin_amdgcn_global_load_lds((g_void*)src, dst, 16, 0, 0);
  }
}

template <int COLS>
__device__ __forceinline__ short8 ld8_swz(const bf16* lds_tile, int row,
                                          int colb) {
  return *reinterpret_cast<const short8*>(
      reinterpret_cast<const char*>(lds_tile) + row * (COLS * 2) +
      (colb ^ ((row & 7) << 4)));
}

template <int COLS>
__device__ __forceinline__ void st16_swz(bf16* lds_tile, int row, int col,
                                         bf16 v) {
  *reinterpret_cast<bf16*>(reinterpret_cast<char*>(lds_tile) + row * (COLS * 2) +
                           ((col * 2) ^ ((row & 7) << 4))) = v;
}

// wave-local fence: P/dS LDS writes must land before same-wave ld8 reads
__device__ __forceinline__ void wave_lds_fence() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
}

// ---------------------------------------------------------------- forward
// Q[B,S,H,D] (strided), K[B,S,Hkv,D] (strided), Vt[B,Hkv,D,S] (contiguous)
// -> O[B,S,H,D] (contiguous), lse[B,S,H] fp32.
template <int D, int NW>  // NW waves x 16 q-rows per block
__global__ __launch_bounds__(512) void flash_fwd_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ Vt, bf16* __restrict__ O, float* __restrict__ lse,
    int B, int H, int Hkv, int S, float scale, int causal,
    int64_t sqb, int64_t sqs, int64_t sqh,
    int64_t skb, int64_t sks, int64_t skh) {
  constexpr int DS = D / 32;
  constexpr int DT = D / 16;
  constexpr int QT = NW * 16;  // q rows per block
  // NW=8: one K/Vt stage feeds 2x the compute of the 4-wave version
  // (halves staging traffic, +50% waves/CU at 48 KB LDS)
  const int n_qt = S / QT;
  const int bh = blockIdx.x / n_qt;
  // longest-trip q-tiles first: causal trip count grows with qt
  const int qt = n_qt - 1 - (blockIdx.x - bh * n_qt);
  const int b = bh / H, h = bh - b * H;
  const int hkv = h / (H / Hkv);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  const int q0 = qt * QT + wid * 16;

  const bf16* Qb = Q + b * sqb + h * sqh;
  const bf16* Kb = K + b * skb + hkv * skh;
  const bf16* Vtb = Vt + ((int64_t)(b * Hkv + hkv) * D) * S;

  __shared__ bf16 k_lds[64 * D];
  __shared__ bf16 vt_lds[D * 64];
  __shared__ bf16 p_lds_all[NW][16 * 64];
  bf16* p_lds = p_lds_all[wid];

  short8 qf[DS];
#pragma unroll
  for (int ds = 0; ds < DS; ++ds)
    qf[ds] = ld8(Qb + (int64_t)(q0 + li) * sqs + ds * 32 + lg * 8);

  float m[4], l[4];
  f32x4 o_acc[DT];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m[r] = NEG_INF; l[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) o_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? (qt * QT + QT) : S;
  for (int kv = 0; kv < kv_end; kv += 64) {
    __syncthreads();  // all waves done reading the previous tile
    stage_tile<64, D, NW * 64>(Kb + (int64_t)kv * sks, sks, k_lds, threadIdx.x);
    // Vt tile: rows d (stride S), cols k in [kv, kv+64)
    stage_tile<D, 64, NW * 64>(Vtb + kv, S, vt_lds, threadIdx.x);
    __syncthreads();  // staged (syncthreads drains vmcnt)

    // ---- S = scale * Q K^T
    f32x4 s[4];
    FWD_PRIO(1);
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 acc{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ds = 0; ds < DS; ++ds)
        acc = mfma16(qf[ds], ld8_swz<D>(k_lds, sub * 16 + li, ds * 64 + lg * 16), acc);
      s[sub] = acc;
    }
    FWD_PRIO(0);
    // ---- online softmax per q-row (reg r), row owned by 16-lane group
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
      // defer-max (guide T13): if this tile's max does not exceed the
      // running max by more than THR, keep the old max — P values stay
      // bounded by e^THR (fp32 accumulates fine) and the O-wide rescale
      // pass is skipped (alpha == 1)
// A/B on MI355X (3x interleaved): defer-max = 230 TF vs plain 240 TF —
// our per-tile rescale is only 32 muls/lane, cheaper than the added
// branches. Off by default; kept for structures with wider O state.
#ifndef PRIME_DEFER_MAX
#define PRIME_DEFER_MAX 0
#endif
#if PRIME_DEFER_MAX
      const float THR = 8.f;
      float mnew = m[r];
      if (tmax > m[r] + 0.f) mnew = (tmax - m[r] <= THR && m[r] > NEG_INF)
                                        ? m[r] : fmaxf(m[r], tmax);
#else
      const float mnew = fmaxf(m[r], tmax);
#endif
      alpha[r] = (mnew == m[r]) ? 1.f : __expf(m[r] - mnew);
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
    float amin = 1.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) amin = fminf(amin, alpha[r]);
    if (__ballot(amin < 1.f)) {
#pragma unroll
      for (int dt = 0; dt < DT; ++dt)
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[dt][r] *= alpha[r];
    }
    // ---- P (C-layout) -> swizzled per-wave LDS (A-layout source)
#pragma unroll
    for (int sub = 0; sub < 4; ++sub)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        st16_swz<64>(p_lds, lg * 4 + r, sub * 16 + li, f2bf(s[sub][r]));
    wave_lds_fence();
    // ---- O += P V
    FWD_PRIO(1);
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        o_acc[dt] = mfma16(ld8_swz<64>(p_lds, li, ks * 64 + lg * 16),
                           ld8_swz<64>(vt_lds, dt * 16 + li, ks * 64 + lg * 16),
                           o_acc[dt]);
    FWD_PRIO(0);
  }
  // ---- epilogue
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv = (l[r] > 0.f) ? 1.f / l[r] : 0.f;
    const int qg = q0 + lg * 4 + r;
    bf16* orow = O + (((int64_t)(b * S + qg)) * H + h) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
      orow[dt * 16 + li] = f2bf(o_acc[dt][r] * inv);
    if (li == 0)
      lse[((int64_t)(b * S + qg)) * H + h] = m[r] + __logf(fmaxf(l[r], 1e-30f));
  }
}


// ------------------------------------------------- forward v5 (32x32 MFMA)
// 8-wave swapped-operand structure (guide §B attn, 8-warp 32x32 ladder):
// St = K·Q^T with v_mfma_f32_32x32x16_bf16 puts each q-row's scores
// LANE-LOCAL (lane pair (l, l^32) splits the row), so the online softmax
// is fully in-register and P feeds the PV A-fragment after one lane-pair
// exchange (v_cvt_pk_bf16_f32 pack + v_permlane32_swap_b32 — guide T12).
// 8 waves x 32 q-rows = 256-row q tiles sharing one K/Vt stage (vs the
// 16x16 kernel's per-wave P LDS bounce + 4x staging traffic). The round-1
// v4 (4-wave, shfl-based exchange, always-rescale) measured 130 TF; the
// deltas here are the rest of the technique stack: defer-max rescale
// skipping (T13), cheap P exchange, per-wave causal trip clipping, and
// setprio around the MFMA clusters (T5).
//
// 32x32x16 fragment maps (HW-verified, test_mfma32_layout_vs_matmul):
//   A[32][16]: lane holds A[lane&31][(lane>>5)*8 + j]
//   B[16][32]: lane holds B[(lane>>5)*8 + j][lane&31]
//   C[32][32]: lane holds C[(r&3) + 8*(r>>2) + 4*(lane>>5)][lane&31]
typedef __attribute__((ext_vector_type(16))) float f32x16;

__device__ __forceinline__ f32x16 mfma32(short8 a, short8 b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

// pack two f32 into one dword of 2 bf16 (no builtin on gfx950 — guide T12)
__device__ __forceinline__ unsigned cvt_pk_bf16(float lo_, float hi_) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo_), "v"(hi_));
  return r;
}

// v_permlane32_swap_b32 a, b: a's lanes>=32 swap with b's lanes<32 —
// after the swap, lane l<32 sees (a_l, a_{l+32}) and lane l>=32 sees
// (b_{l-32}, b_l) in (a, b)
__device__ __forceinline__ void permlane32_swap(unsigned& a, unsigned& b) {
  asm volatile("v_permlane32_swap_b32 %0, %1" : "+v"(a), "+v"(b));
}

// FLAGS bits: 1 = setprio around MFMA clusters, 2 = defer-max (THR=8),
// 4 = 3-buffer deep prefetch with counted vmcnt at the tile boundary
// (guide T4: __syncthreads' vmcnt(0) drain waits on the JUST-issued
// stage; with 2 tiles of lookahead the boundary only needs the oldest
// in-flight tile, so vmcnt(4) leaves the newest 4 loads flying)
template <int D, int FLAGS>
__global__ __launch_bounds__(512, 2) void flash_fwd_v5_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ Vt, bf16* __restrict__ O, float* __restrict__ lse,
    int B, int H, int Hkv, int S, float scale, int causal,
    int64_t sqb, int64_t sqs, int64_t sqh,
    int64_t skb, int64_t sks, int64_t skh) {
  constexpr int DSL = D / 16;   // 16-wide d slices for the K dim
  constexpr int DT32 = D / 32;  // 32-wide output d tiles
  const int n_qt = S / 256;
  // XCD-aware bijective remap (guide T1): consecutive qt-tiles of one
  // (b,h) land on ONE XCD's L2, which then reuses the same K/V stream
  const int nwg = gridDim.x;
  int vb = blockIdx.x;
  if ((nwg & 7) == 0) vb = (blockIdx.x & 7) * (nwg >> 3) + (blockIdx.x >> 3);
  const int bh = vb / n_qt;
  const int qt = n_qt - 1 - (vb - bh * n_qt);  // longest trips first
  const int b = bh / H, h = bh - b * H;
  const int hkv = h / (H / Hkv);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 31, hi = lane >> 5;
  const int q0w = qt * 256 + wid * 32;  // this wave's first q row

  const bf16* Qb = Q + b * sqb + h * sqh;
  const bf16* Kb = K + b * skb + hkv * skh;
  const bf16* Vtb = Vt + ((int64_t)(b * Hkv + hkv) * D) * S;

  constexpr int NBUF = (FLAGS & 4) ? 3 : 2;
  __shared__ bf16 k_lds[NBUF][64 * D];
  __shared__ bf16 vt_lds[NBUF][D * 64];
  __shared__ float bcast_all[8][32];  // per-wave alpha / inv-l broadcast
  float* bcast = bcast_all[wid];

  // Q^T B-fragments, pre-scaled: lane holds Q[q0w+lo][ds*16+hi*8+j]
  short8 qf[DSL];
#pragma unroll
  for (int ds = 0; ds < DSL; ++ds) {
    const bf16* src = Qb + (int64_t)(q0w + lo) * sqs + ds * 16 + hi * 8;
    short8 raw = ld8(src);
    short8 sc;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bf16 v = reinterpret_cast<const bf16*>(&raw)[j];
      reinterpret_cast<bf16*>(&sc)[j] = f2bf(bf2f(v) * scale);
    }
    qf[ds] = sc;
  }

  float m_run = NEG_INF, l_run = 0.f;
  f32x16 o_acc[DT32];
#pragma unroll
  for (int dt = 0; dt < DT32; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[dt][r] = 0.f;

  const int kv_end = causal ? (qt * 256 + 256) : S;
  // per-wave causal clip: this wave's rows end at q0w+31, so tiles past
  // q0w+32 are fully masked — skip their compute (they still stage+sync)
  const int my_kv_end = causal ? (q0w + 32) : S;
  stage_tile<64, D, 512>(Kb, sks, k_lds[0], threadIdx.x);
  stage_tile<D, 64, 512>(Vtb, S, vt_lds[0], threadIdx.x);
  if (FLAGS & 4) {  // 2-deep prologue
    if (64 < kv_end) {
      stage_tile<64, D, 512>(Kb + (int64_t)64 * sks, sks, k_lds[1], threadIdx.x);
      stage_tile<D, 64, 512>(Vtb + 64, S, vt_lds[1], threadIdx.x);
    }
  }
  int idx = 0;
  for (int kv = 0, kt = 0; kv < kv_end; kv += 64, ++kt) {
    idx = kt % NBUF;
    if (FLAGS & 4) {
      // tile kv's stages are >=1 full tile old: only the newest tile's
      // 4 loads may still fly across this rendezvous
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_sched_barrier(0);
      const int pre = kv + 2 * 64;
      if (pre < kv_end) {
        stage_tile<64, D, 512>(Kb + (int64_t)pre * sks, sks,
                               k_lds[(kt + 2) % NBUF], threadIdx.x);
        stage_tile<D, 64, 512>(Vtb + pre, S, vt_lds[(kt + 2) % NBUF],
                               threadIdx.x);
      }
    } else {
      __syncthreads();  // prev tile consumed + this tile's stage drained
      if (kv + 64 < kv_end) {
        // prefetch-behind-barrier: these loads get the whole tile's
        // compute to land (the next syncthreads drains them)
        stage_tile<64, D, 512>(Kb + (int64_t)(kv + 64) * sks, sks,
                               k_lds[(kt + 1) % NBUF], threadIdx.x);
        stage_tile<D, 64, 512>(Vtb + kv + 64, S, vt_lds[(kt + 1) % NBUF],
                               threadIdx.x);
      }
    }
    if (kv >= my_kv_end) continue;  // fully-masked tile for this wave

    // ---- St = (K q^T): two 32x32 C tiles over the 64-key block
    f32x16 st[2];
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int r = 0; r < 16; ++r) st[t][r] = 0.f;
#pragma unroll
      for (int ds = 0; ds < DSL; ++ds) {
        const short8 kf = ld8_swz<D>(k_lds[idx], t * 32 + lo, ds * 32 + hi * 16);
        st[t] = mfma32(kf, qf[ds], st[t]);
      }
    }
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(0);
    // ---- online softmax: st[t][r] is k = kv + t*32 + crow(r,hi) for
    // q row q0w+lo; lane pair (l, l^32) splits the row's 64 scores
    float tmax = NEG_INF;
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg = kv + t * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        if (causal && kg > q0w + lo) st[t][r] = NEG_INF;
        tmax = fmaxf(tmax, st[t][r]);
      }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    // defer-max (guide T13): skip the O-wide rescale while the running
    // max still bounds P by e^THR (fp32 accumulation tolerates it)
    const float thr = (FLAGS & 2) ? 8.f : 0.f;
    float alpha = 1.f;
    if (__ballot(tmax > m_run + thr)) {
      const float mnew = fmaxf(m_run, tmax);
      alpha = __expf(m_run - mnew);  // 1 for rows that did not grow
      m_run = mnew;
      if (hi == 0) bcast[lo] = alpha;
      wave_lds_fence();
#pragma unroll
      for (int dt = 0; dt < DT32; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r)
          o_acc[dt][r] *= bcast[(r & 3) + 8 * (r >> 2) + 4 * hi];
    }
    float psum = 0.f;
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float pv = (st[t][r] <= NEG_INF) ? 0.f : __expf(st[t][r] - m_run);
        st[t][r] = pv;
        psum += pv;
      }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * alpha + psum;

    // ---- P -> PV A-fragments (guide T12): per 16-k slice, pack the two
    // 4-value r-groups to bf16 dwords and permlane32_swap — lane l<32
    // keeps x and receives partner's x; lane>=32 receives partner's y and
    // keeps y. Result is branch-free and identical on both halves.
    short8 pa[2 * 2];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int kst = 0; kst < 2; ++kst) {
        const int g0 = 4 * (2 * kst), g1 = 4 * (2 * kst + 1);
        unsigned x0 = cvt_pk_bf16(st[t][g0 + 0], st[t][g0 + 1]);
        unsigned x1 = cvt_pk_bf16(st[t][g0 + 2], st[t][g0 + 3]);
        unsigned y0 = cvt_pk_bf16(st[t][g1 + 0], st[t][g1 + 1]);
        unsigned y1 = cvt_pk_bf16(st[t][g1 + 2], st[t][g1 + 3]);
        permlane32_swap(x0, y0);
        permlane32_swap(x1, y1);
        short8 frag;
        reinterpret_cast<unsigned*>(&frag)[0] = x0;
        reinterpret_cast<unsigned*>(&frag)[1] = x1;
        reinterpret_cast<unsigned*>(&frag)[2] = y0;
        reinterpret_cast<unsigned*>(&frag)[3] = y1;
        pa[t * 2 + kst] = frag;
      }
    }
    // ---- O += P V  (B-frags from swizzled Vt LDS)
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < DT32; ++dt)
#pragma unroll
      for (int ks = 0; ks < 4; ++ks)
        o_acc[dt] = mfma32(pa[ks],
                           ld8_swz<64>(vt_lds[idx], dt * 32 + lo, ks * 32 + hi * 16),
                           o_acc[dt]);
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(0);
  }
  // ---- epilogue: O /= l (per-row broadcast), write + lse
  if (hi == 0) bcast[lo] = (l_run > 0.f) ? 1.f / l_run : 0.f;
  wave_lds_fence();
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qr = (r & 3) + 8 * (r >> 2) + 4 * hi;
    const float inv = bcast[qr];
    bf16* orow = O + (((int64_t)(b * S + q0w + qr)) * H + h) * D;
#pragma unroll
    for (int dt = 0; dt < DT32; ++dt)
      orow[dt * 32 + lo] = f2bf(o_acc[dt][r] * inv);
  }
  if (hi == 0)
    lse[((int64_t)(b * S + q0w + lo)) * H + h] = m_run + __logf(fmaxf(l_run, 1e-30f));
}

// ------------------------------------------------------- delta = rowsum(dO*O)
// dO, O: contiguous [B,S,H,D] -> delta [B,S,H] (row-ordered, contiguous)
__global__ void attn_delta_kernel(const bf16* __restrict__ dO,
                                  const bf16* __restrict__ O,
                                  float* __restrict__ delta, int64_t rows,
                                  int D) {
  const int64_t row0 = (int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  for (int64_t r = row0; r < rows; r += (int64_t)gridDim.x * (blockDim.x >> 6)) {
    float acc = 0.f;
    for (int i = lane * 2; i < D; i += 128) {
      acc += bf2f(dO[r * D + i]) * bf2f(O[r * D + i]);
      acc += bf2f(dO[r * D + i + 1]) * bf2f(O[r * D + i + 1]);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) delta[r] = acc;
  }
}

// ------------------------------------------------ backward dQ v5 (32x32)
// Same 8-wave swapped-operand structure as the v5 forward: St = K·Q^T and
// dPt = V·dO^T give lane-local q-rows (q = lane&31), so lse/delta are one
// scalar per lane and dS never touches LDS — it is packed straight into
// the dQ-accumulation A-fragments with cvt_pk + permlane32_swap.
template <int D, int FLAGS>
__global__ __launch_bounds__(512, 2) void flash_bwd_dq_v5_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ Kt,
    const bf16* __restrict__ dO, const float* __restrict__ lse,
    const float* __restrict__ delta, bf16* __restrict__ dQ, int B, int H,
    int Hkv, int S, float scale, int causal,
    int64_t sqb, int64_t sqs, int64_t sqh,
    int64_t skb, int64_t sks, int64_t skh,
    int64_t svb, int64_t svs, int64_t svh) {
  constexpr int DSL = D / 16;
  constexpr int DT32 = D / 32;
  const int n_qt = S / 256;
  const int nwg = gridDim.x;
  int vb = blockIdx.x;
  if ((nwg & 7) == 0) vb = (blockIdx.x & 7) * (nwg >> 3) + (blockIdx.x >> 3);
  const int bh = vb / n_qt;
  const int qt = n_qt - 1 - (vb - bh * n_qt);
  const int b = bh / H, h = bh - b * H;
  const int hkv = h / (H / Hkv);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 31, hi = lane >> 5;
  const int q0w = qt * 256 + wid * 32;

  const bf16* Qb = Q + b * sqb + h * sqh;
  const bf16* Kb = K + b * skb + hkv * skh;
  const bf16* Vb = V + b * svb + hkv * svh;
  const bf16* Ktb = Kt + ((int64_t)(b * Hkv + hkv) * D) * S;
  const bf16* dOb = dO + ((int64_t)b * S * H + h) * D;  // row stride H*D
  const float* lse_b = lse + (int64_t)b * S * H + h;    // stride H
  const float* dl_b = delta + (int64_t)b * S * H + h;

  __shared__ bf16 k_lds[2][64 * D];
  __shared__ bf16 v_lds[2][64 * D];
  __shared__ bf16 kt_lds[2][D * 64];

  // Q^T / dO^T B-fragments for this wave's 32 q rows (q = lo)
  short8 qf[DSL], dof[DSL];
#pragma unroll
  for (int ds = 0; ds < DSL; ++ds) {
    const bf16* src = Qb + (int64_t)(q0w + lo) * sqs + ds * 16 + hi * 8;
    short8 raw = ld8(src);
    short8 sc;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bf16 v = reinterpret_cast<const bf16*>(&raw)[j];
      reinterpret_cast<bf16*>(&sc)[j] = f2bf(bf2f(v) * scale);
    }
    qf[ds] = sc;
    dof[ds] = ld8(dOb + (int64_t)(q0w + lo) * H * D + ds * 16 + hi * 8);
  }
  const float lse_q = lse_b[(int64_t)(q0w + lo) * H];
  const float dl_q = dl_b[(int64_t)(q0w + lo) * H];

  f32x16 dq_acc[DT32];
#pragma unroll
  for (int dt = 0; dt < DT32; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[dt][r] = 0.f;

  const int kv_end = causal ? (qt * 256 + 256) : S;
  const int my_kv_end = causal ? (q0w + 32) : S;
  stage_tile<64, D, 512>(Kb, sks, k_lds[0], threadIdx.x);
  stage_tile<64, D, 512>(Vb, svs, v_lds[0], threadIdx.x);
  stage_tile<D, 64, 512>(Ktb, S, kt_lds[0], threadIdx.x);
  int idx = 0;
  for (int kv = 0; kv < kv_end; kv += 64, idx ^= 1) {
    __syncthreads();
    if (kv + 64 < kv_end) {
      stage_tile<64, D, 512>(Kb + (int64_t)(kv + 64) * sks, sks,
                             k_lds[idx ^ 1], threadIdx.x);
      stage_tile<64, D, 512>(Vb + (int64_t)(kv + 64) * svs, svs,
                             v_lds[idx ^ 1], threadIdx.x);
      stage_tile<D, 64, 512>(Ktb + kv + 64, S, kt_lds[idx ^ 1], threadIdx.x);
    }
    if (kv >= my_kv_end) continue;

    // ---- St = K·Q^T (pre-scaled), dPt = V·dO^T: C[k][q], q = lo
    f32x16 st[2], dpt[2];
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int r = 0; r < 16; ++r) { st[t][r] = 0.f; dpt[t][r] = 0.f; }
#pragma unroll
      for (int ds = 0; ds < DSL; ++ds) {
        const short8 kf = ld8_swz<D>(k_lds[idx], t * 32 + lo, ds * 32 + hi * 16);
        const short8 vf = ld8_swz<D>(v_lds[idx], t * 32 + lo, ds * 32 + hi * 16);
        st[t] = mfma32(kf, qf[ds], st[t]);
        dpt[t] = mfma32(vf, dof[ds], dpt[t]);
      }
    }
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(0);
    // ---- dS = P * (dP - delta) * scale, all in-register (q = lo)
#pragma unroll
    for (int t = 0; t < 2; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg = kv + t * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float p = __expf(st[t][r] - lse_q);
        if (causal && kg > q0w + lo) p = 0.f;
        st[t][r] = p * (dpt[t][r] - dl_q) * scale;
      }
    // ---- dS -> A-fragments (same lane-pair exchange as the fwd P)
    short8 da[2 * 2];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
#pragma unroll
      for (int kst = 0; kst < 2; ++kst) {
        const int g0 = 4 * (2 * kst), g1 = 4 * (2 * kst + 1);
        unsigned x0 = cvt_pk_bf16(st[t][g0 + 0], st[t][g0 + 1]);
        unsigned x1 = cvt_pk_bf16(st[t][g0 + 2], st[t][g0 + 3]);
        unsigned y0 = cvt_pk_bf16(st[t][g1 + 0], st[t][g1 + 1]);
        unsigned y1 = cvt_pk_bf16(st[t][g1 + 2], st[t][g1 + 3]);
        permlane32_swap(x0, y0);
        permlane32_swap(x1, y1);
        short8 frag;
        reinterpret_cast<unsigned*>(&frag)[0] = x0;
        reinterpret_cast<unsigned*>(&frag)[1] = x1;
        reinterpret_cast<unsigned*>(&frag)[2] = y0;
        reinterpret_cast<unsigned*>(&frag)[3] = y1;
        da[t * 2 + kst] = frag;
      }
    }
    // ---- dQ += dS·K (B-frags from swizzled Kt LDS)
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < DT32; ++dt)
#pragma unroll
      for (int ks = 0; ks < 4; ++ks)
        dq_acc[dt] = mfma32(da[ks],
                            ld8_swz<64>(kt_lds[idx], dt * 32 + lo, ks * 32 + hi * 16),
                            dq_acc[dt]);
    if (FLAGS & 1) __builtin_amdgcn_s_setprio(0);
  }
  // ---- epilogue: C[q][d] rows q = crow(r,hi)
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qr = (r & 3) + 8 * (r >> 2) + 4 * hi;
    bf16* row = dQ + (((int64_t)(b * S + q0w + qr)) * H + h) * D;
#pragma unroll
    for (int dt = 0; dt < DT32; ++dt)
      row[dt * 32 + lo] = f2bf(dq_acc[dt][r]);
  }
}

// ----------------------------------------------------------- backward dQ
// Stages K[64][D], V[64][D] (B-operands for S and dP) and Kt[D][64]
// (B-operand for dQ += dS*K). dO is contiguous [B,S,H,D].
template <int D>
__global__ __launch_bounds__(256) void flash_bwd_dq_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ Kt,
    const bf16* __restrict__ dO, const float* __restrict__ lse,
    const float* __restrict__ delta, bf16* __restrict__ dQ, int B, int H,
    int Hkv, int S, float scale, int causal,
    int64_t sqb, int64_t sqs, int64_t sqh,
    int64_t skb, int64_t sks, int64_t skh,
    int64_t svb, int64_t svs, int64_t svh) {
  constexpr int DS = D / 32;
  constexpr int DT = D / 16;
  const int n_qt = S / 64;
  const int bh = blockIdx.x / n_qt;
  const int qt = n_qt - 1 - (blockIdx.x - bh * n_qt);  // longest trips first
  const int b = bh / H, h = bh - b * H;
  const int hkv = h / (H / Hkv);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  const int q0 = qt * 64 + wid * 16;

  const bf16* Qb = Q + b * sqb + h * sqh;
  const bf16* Kb = K + b * skb + hkv * skh;
  const bf16* Vb = V + b * svb + hkv * svh;
  const bf16* Ktb = Kt + ((int64_t)(b * Hkv + hkv) * D) * S;
  const bf16* dOb = dO + ((int64_t)b * S * H + h) * D;  // row stride H*D
  const float* lse_b = lse + (int64_t)b * S * H + h;    // stride H
  const float* dl_b = delta + (int64_t)b * S * H + h;

  __shared__ bf16 k_lds[64 * D];
  __shared__ bf16 v_lds[64 * D];
  __shared__ bf16 kt_lds[D * 64];
  __shared__ bf16 ds_lds_all[4][16 * 64];
  bf16* ds_lds = ds_lds_all[wid];

  short8 qf[DS], dof[DS];
#pragma unroll
  for (int ds = 0; ds < DS; ++ds) {
    qf[ds] = ld8(Qb + (int64_t)(q0 + li) * sqs + ds * 32 + lg * 8);
    dof[ds] = ld8(dOb + (int64_t)(q0 + li) * H * D + ds * 32 + lg * 8);
  }
  float lse_r[4], dl_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    lse_r[r] = lse_b[(int64_t)(q0 + lg * 4 + r) * H];
    dl_r[r] = dl_b[(int64_t)(q0 + lg * 4 + r) * H];
  }
  f32x4 dq_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) dq_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? (qt * 64 + 64) : S;
  for (int kv = 0; kv < kv_end; kv += 64) {
    __syncthreads();
    stage_tile<64, D>(Kb + (int64_t)kv * sks, sks, k_lds, threadIdx.x);
    stage_tile<64, D>(Vb + (int64_t)kv * svs, svs, v_lds, threadIdx.x);
    stage_tile<D, 64>(Ktb + kv, S, kt_lds, threadIdx.x);
    __syncthreads();
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      f32x4 s_acc{0.f, 0.f, 0.f, 0.f}, dp_acc{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ds = 0; ds < DS; ++ds) {
        s_acc = mfma16(qf[ds], ld8_swz<D>(k_lds, sub * 16 + li, ds * 64 + lg * 16), s_acc);
        dp_acc = mfma16(dof[ds], ld8_swz<D>(v_lds, sub * 16 + li, ds * 64 + lg * 16), dp_acc);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qg = q0 + lg * 4 + r;
        const int kg = kv + sub * 16 + li;
        float p = __expf(s_acc[r] * scale - lse_r[r]);
        if (causal && kg > qg) p = 0.f;
        st16_swz<64>(ds_lds, lg * 4 + r, sub * 16 + li,
                     f2bf(p * (dp_acc[r] - dl_r[r]) * scale));
      }
    }
    wave_lds_fence();
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        dq_acc[dt] = mfma16(ld8_swz<64>(ds_lds, li, ks * 64 + lg * 16),
                            ld8_swz<64>(kt_lds, dt * 16 + li, ks * 64 + lg * 16),
                            dq_acc[dt]);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    bf16* row = dQ + (((int64_t)(b * S + q0 + lg * 4 + r)) * H + h) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) row[dt * 16 + li] = f2bf(dq_acc[dt][r]);
  }
}

// ---------------------------------------------- backward dK/dV v5 (32x32)
// Swapped the other way around from the fwd/dq kernels: St^T = Q·K^T puts
// each lane's scores at a FIXED key column (k = lane&31, the dimension
// this kernel accumulates over), so P^T/dS^T feed the dV/dK accumulation
// A-fragments after the same cvt_pk + permlane32_swap lane-pair exchange
// — no per-wave P/dS LDS bounce. K/V rows live in registers (per-wave
// constants), Q/dO rows are read per-lane from global (L1-resident across
// the d-slice loop), and only the genuinely-transposed operands (Qt/dOt
// tiles for the accumulation B-fragments) are staged, double-buffered.
// 8 waves x 32 k-rows = 256-row k tiles; fp32 split workspaces + reduce
// as in the 16x16 kernel.
template <int D, int FLAGS>
__global__ __launch_bounds__(512, 2) void flash_bwd_dkv_v5_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ Qt,
    const bf16* __restrict__ K, const bf16* __restrict__ V,
    const bf16* __restrict__ dO, const bf16* __restrict__ dOt,
    const float* __restrict__ lse, const float* __restrict__ delta,
    float* __restrict__ wsK, float* __restrict__ wsV, int B, int H, int Hkv,
    int S, float scale, int causal, int splits,
    int64_t sqb, int64_t sqs, int64_t sqh,
    int64_t skb, int64_t sks, int64_t skh,
    int64_t svb, int64_t svs, int64_t svh) {
  constexpr int DSL = D / 16;
  constexpr int DT32 = D / 32;
  const int n_kt = S / 256;
  const int per_split = (B * Hkv) * n_kt;
  const int split = blockIdx.x / per_split;
  const int rem = blockIdx.x - split * per_split;
  const int bh = rem / n_kt;
  const int kt = rem - bh * n_kt;
  const int b = bh / Hkv, hkv = bh - b * Hkv;
  const int group = H / Hkv;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 31, hi = lane >> 5;
  const int k0w = kt * 256 + wid * 32;  // this wave's first k row (global)

  const bf16* Kb = K + b * skb + hkv * skh;
  const bf16* Vb = V + b * svb + hkv * svh;

  __shared__ bf16 q_lds[2][64 * D];
  __shared__ bf16 do_lds[2][64 * D];
  __shared__ bf16 qt_lds[2][D * 64];
  __shared__ bf16 dot_lds[2][D * 64];

  // K^T / V^T B-fragment source rows (k0w+lo): loaded inline per MFMA —
  // holding all 8 slices of both in registers (64 VGPRs) pushed the
  // kernel to 256 regs + scratch spills; the rows stay L1-resident
  // across the q loop instead
  const bf16* krow = Kb + (int64_t)(k0w + lo) * sks;
  const bf16* vrow = Vb + (int64_t)(k0w + lo) * svs;
  f32x16 dk_acc[DT32], dv_acc[DT32];
#pragma unroll
  for (int dt = 0; dt < DT32; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) { dk_acc[dt][r] = 0.f; dv_acc[dt][r] = 0.f; }

  // flattened (g, q-tile) loop for double-buffered Qt/dOt prefetch
  const int q_start = (causal ? kt * 256 : 0) + split * 64;
  const int tiles_per_g = (S - q_start + 64 * splits - 1) / (64 * splits);
  const int n_iter = group * tiles_per_g;
  const auto stage_for = [&](int it, int buf) {
    const int g = it / tiles_per_g;
    const int q0g = q_start + (it - g * tiles_per_g) * 64 * splits;
    const int h = hkv * group + g;
    const bf16* Qb_ = Q + b * sqb + h * sqh;
    const bf16* dOb_ = dO + ((int64_t)b * S * H + h) * D;
    const bf16* Qtb = Qt + ((int64_t)(b * H + h) * D) * S;
    const bf16* dOtb = dOt + ((int64_t)(b * H + h) * D) * S;
    stage_tile<64, D, 512>(Qb_ + (int64_t)q0g * sqs, sqs, q_lds[buf], threadIdx.x);
    stage_tile<64, D, 512>(dOb_ + (int64_t)q0g * H * D, (int64_t)H * D, do_lds[buf], threadIdx.x);
    stage_tile<D, 64, 512>(Qtb + q0g, S, qt_lds[buf], threadIdx.x);
    stage_tile<D, 64, 512>(dOtb + q0g, S, dot_lds[buf], threadIdx.x);
  };
  stage_for(0, 0);
  for (int it = 0; it < n_iter; ++it) {
    const int cur = it & 1;
    __syncthreads();
    if (it + 1 < n_iter)
      stage_for(it + 1, cur ^ 1);
    const int g = it / tiles_per_g;
    const int q0g = q_start + (it - g * tiles_per_g) * 64 * splits;
    if (causal && q0g + 64 <= k0w) continue;  // fully masked for this wave
    const int h = hkv * group + g;
    const float* lse_b = lse + (int64_t)b * S * H + h;
    const float* dl_b = delta + (int64_t)b * S * H + h;

#pragma unroll
    for (int qs = 0; qs < 2; ++qs) {
      const int q32 = q0g + qs * 32;
      // St^T = Q·K^T, dPt^T = dO·V^T: C[32q][32k], k = lo (lane-local)
      f32x16 st, dpt;
#pragma unroll
      for (int r = 0; r < 16; ++r) { st[r] = 0.f; dpt[r] = 0.f; }
      if (FLAGS & 1) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ds = 0; ds < DSL; ++ds) {
        const short8 qfr = ld8_swz<D>(q_lds[cur], qs * 32 + lo, ds * 32 + hi * 16);
        st = mfma32(qfr, ld8(krow + ds * 16 + hi * 8), st);
      }
#pragma unroll
      for (int ds = 0; ds < DSL; ++ds) {
        const short8 dofr = ld8_swz<D>(do_lds[cur], qs * 32 + lo, ds * 32 + hi * 16);
        dpt = mfma32(dofr, ld8(vrow + ds * 16 + hi * 8), dpt);
      }
      if (FLAGS & 1) __builtin_amdgcn_s_setprio(0);
      // P^T / dS^T in-register; lse/delta per q-row crow(r,hi) are
      // broadcast loads (same address across the 32 lo-lanes)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qg = q32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float lse_q = lse_b[(int64_t)qg * H];
        const float dl_q = dl_b[(int64_t)qg * H];
        float p = __expf(st[r] * scale - lse_q);
        if (causal && (k0w + lo) > qg) p = 0.f;
        st[r] = p;
        dpt[r] = p * (dpt[r] - dl_q) * scale;
      }
      // pack P^T / dS^T into accumulation A-fragments: A[32k][16q], the
      // q-slices split across the lane pair exactly like the fwd P
      short8 pa[2], da[2];
#pragma unroll
      for (int kst = 0; kst < 2; ++kst) {
        const int g0 = 4 * (2 * kst), g1 = 4 * (2 * kst + 1);
        unsigned x0 = cvt_pk_bf16(st[g0 + 0], st[g0 + 1]);
        unsigned x1 = cvt_pk_bf16(st[g0 + 2], st[g0 + 3]);
        unsigned y0 = cvt_pk_bf16(st[g1 + 0], st[g1 + 1]);
        unsigned y1 = cvt_pk_bf16(st[g1 + 2], st[g1 + 3]);
        permlane32_swap(x0, y0);
        permlane32_swap(x1, y1);
        short8 f;
        reinterpret_cast<unsigned*>(&f)[0] = x0;
        reinterpret_cast<unsigned*>(&f)[1] = x1;
        reinterpret_cast<unsigned*>(&f)[2] = y0;
        reinterpret_cast<unsigned*>(&f)[3] = y1;
        pa[kst] = f;
        x0 = cvt_pk_bf16(dpt[g0 + 0], dpt[g0 + 1]);
        x1 = cvt_pk_bf16(dpt[g0 + 2], dpt[g0 + 3]);
        y0 = cvt_pk_bf16(dpt[g1 + 0], dpt[g1 + 1]);
        y1 = cvt_pk_bf16(dpt[g1 + 2], dpt[g1 + 3]);
        permlane32_swap(x0, y0);
        permlane32_swap(x1, y1);
        reinterpret_cast<unsigned*>(&f)[0] = x0;
        reinterpret_cast<unsigned*>(&f)[1] = x1;
        reinterpret_cast<unsigned*>(&f)[2] = y0;
        reinterpret_cast<unsigned*>(&f)[3] = y1;
        da[kst] = f;
      }
      // dV += P^T·dO, dK += dS^T·Q (B-frags from swizzled dOt/Qt tiles)
      if (FLAGS & 1) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < DT32; ++dt)
#pragma unroll
        for (int kst = 0; kst < 2; ++kst) {
          const int colb = qs * 64 + kst * 32 + hi * 16;
          dv_acc[dt] = mfma32(pa[kst],
                              ld8_swz<64>(dot_lds[cur], dt * 32 + lo, colb),
                              dv_acc[dt]);
          dk_acc[dt] = mfma32(da[kst],
                              ld8_swz<64>(qt_lds[cur], dt * 32 + lo, colb),
                              dk_acc[dt]);
        }
      if (FLAGS & 1) __builtin_amdgcn_s_setprio(0);
    }
  }
  // fp32 partials: ws[split][b][hkv][kg][d]; C[32k][32d] rows crow(r,hi)
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kg = k0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
    const int64_t base = ((((int64_t)split * B + b) * Hkv + hkv) * S + kg) * D;
#pragma unroll
    for (int dt = 0; dt < DT32; ++dt) {
      wsK[base + dt * 32 + lo] = dk_acc[dt][r];
      wsV[base + dt * 32 + lo] = dv_acc[dt][r];
    }
  }
}

// -------------------------------------------------------- backward dK, dV
// Stages Q[64][D], dO[64][D] (B-operands for St and dPt), Qt[D][64] and
// dOt[D][64] (B-operands for dK and dV accumulation).
// DBUF=2: double-buffered q/do/qt/dot stages with prefetch-behind-barrier
// (one barrier per q-tile instead of the serial stage-between-two-
// barriers, whose zero-gap vmcnt drain shows as this kernel's 9.0
// wait/busy in the PMC profile). At NW=8 this is exactly the 160 KiB
// LDS limit: 2x4x16 KiB stages + 8x2x2 KiB P/dS wave buffers.
template <int D, int NW = 4, int DBUF = 1>  // NW waves x 16 k-rows
__global__ __launch_bounds__(512) void flash_bwd_dkv_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ Qt,
    const bf16* __restrict__ K, const bf16* __restrict__ V,
    const bf16* __restrict__ dO, const bf16* __restrict__ dOt,
    const float* __restrict__ lse, const float* __restrict__ delta,
    float* __restrict__ wsK, float* __restrict__ wsV, int B, int H, int Hkv,
    int S, float scale, int causal, int splits,
    int64_t sqb, int64_t sqs, int64_t sqh,
    int64_t skb, int64_t sks, int64_t skh,
    int64_t svb, int64_t svs, int64_t svh) {
  constexpr int DS = D / 32;
  constexpr int DT = D / 16;
  const int n_kt = S / (NW * 16);
  const int per_split = (B * Hkv) * n_kt;
  const int split = blockIdx.x / per_split;
  const int rem = blockIdx.x - split * per_split;
  const int bh = rem / n_kt;
  const int kt = rem - bh * n_kt;
  const int b = bh / Hkv, hkv = bh - b * Hkv;
  const int group = H / Hkv;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, li = lane & 15;
  const int k0 = kt * (NW * 16) + wid * 16;

  const bf16* Kb = K + b * skb + hkv * skh;
  const bf16* Vb = V + b * svb + hkv * svh;

  __shared__ bf16 q_lds[DBUF][64 * D];
  __shared__ bf16 do_lds[DBUF][64 * D];
  __shared__ bf16 qt_lds[DBUF][D * 64];
  __shared__ bf16 dot_lds[DBUF][D * 64];
  __shared__ bf16 pt_lds_all[NW][16 * 64];
  __shared__ bf16 dst_lds_all[NW][16 * 64];
  bf16* pt_lds = pt_lds_all[wid];
  bf16* dst_lds = dst_lds_all[wid];

  short8 kf[DS], vf[DS];
#pragma unroll
  for (int ds = 0; ds < DS; ++ds) {
    kf[ds] = ld8(Kb + (int64_t)(k0 + li) * sks + ds * 32 + lg * 8);
    vf[ds] = ld8(Vb + (int64_t)(k0 + li) * svs + ds * 32 + lg * 8);
  }
  f32x4 dk_acc[DT], dv_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) {
    dk_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const int q_start = (causal ? (kt * (NW * 16)) / 64 * 64 : 0) + split * 64;
  const int tiles_per_g =
      (S > q_start) ? (S - q_start + 64 * splits - 1) / (64 * splits) : 0;
  const int n_iter = group * tiles_per_g;
  const auto stage_it = [&](int it, int buf) {
    const int g_ = it / tiles_per_g;
    const int q0g_ = q_start + (it - g_ * tiles_per_g) * 64 * splits;
    const int h_ = hkv * group + g_;
    const bf16* Qb_ = Q + b * sqb + h_ * sqh;
    const bf16* dOb_ = dO + ((int64_t)b * S * H + h_) * D;
    const bf16* Qtb_ = Qt + ((int64_t)(b * H + h_) * D) * S;
    const bf16* dOtb_ = dOt + ((int64_t)(b * H + h_) * D) * S;
    stage_tile<64, D, NW * 64>(Qb_ + (int64_t)q0g_ * sqs, sqs, q_lds[buf], threadIdx.x);
    stage_tile<64, D, NW * 64>(dOb_ + (int64_t)q0g_ * H * D, (int64_t)H * D, do_lds[buf], threadIdx.x);
    stage_tile<D, 64, NW * 64>(Qtb_ + q0g_, S, qt_lds[buf], threadIdx.x);
    stage_tile<D, 64, NW * 64>(dOtb_ + q0g_, S, dot_lds[buf], threadIdx.x);
  };
  if (DBUF == 2 && n_iter > 0) stage_it(0, 0);
  for (int it = 0; it < n_iter; ++it) {
    const int g = it / tiles_per_g;
    const int q0g = q_start + (it - g * tiles_per_g) * 64 * splits;
    const int h = hkv * group + g;
    const float* lse_b = lse + (int64_t)b * S * H + h;
    const float* dl_b = delta + (int64_t)b * S * H + h;
    const int cur = (DBUF == 2) ? (it & 1) : 0;
    {
      __syncthreads();
      if (DBUF == 2) {
        // prefetch-behind-barrier: next tile's loads land under this
        // tile's compute; the next barrier drains them
        if (it + 1 < n_iter) stage_it(it + 1, cur ^ 1);
      } else {
        stage_it(it, 0);
        __syncthreads();
      }
      // per-wave causal clip: this wave's k-rows start at k0, so q-tiles
      // entirely below the diagonal are fully masked — skip their
      // compute (the wave still co-staged and hits the next barrier)
      if (causal && q0g + 64 <= k0) continue;
#pragma unroll
      for (int sub = 0; sub < 4; ++sub) {
        f32x4 st_acc{0.f, 0.f, 0.f, 0.f}, dpt_acc{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ds = 0; ds < DS; ++ds) {
          const short8 qb = ld8_swz<D>(q_lds[cur], sub * 16 + li, ds * 64 + lg * 16);
          const short8 dob = ld8_swz<D>(do_lds[cur], sub * 16 + li, ds * 64 + lg * 16);
          st_acc = mfma16(kf[ds], qb, st_acc);
          dpt_acc = mfma16(vf[ds], dob, dpt_acc);
        }
        const int qg = q0g + sub * 16 + li;
        const float lse_q = lse_b[(int64_t)qg * H];
        const float dl_q = dl_b[(int64_t)qg * H];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kg = k0 + lg * 4 + r;
          float p = __expf(st_acc[r] * scale - lse_q);
          if (causal && kg > qg) p = 0.f;
          st16_swz<64>(pt_lds, lg * 4 + r, sub * 16 + li, f2bf(p));
          st16_swz<64>(dst_lds, lg * 4 + r, sub * 16 + li,
                       f2bf(p * (dpt_acc[r] - dl_q) * scale));
        }
      }
      wave_lds_fence();
#pragma unroll
      for (int dt = 0; dt < DT; ++dt)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const short8 pa = ld8_swz<64>(pt_lds, li, ks * 64 + lg * 16);
          const short8 da = ld8_swz<64>(dst_lds, li, ks * 64 + lg * 16);
          dv_acc[dt] = mfma16(pa, ld8_swz<64>(dot_lds[cur], dt * 16 + li, ks * 64 + lg * 16), dv_acc[dt]);
          dk_acc[dt] = mfma16(da, ld8_swz<64>(qt_lds[cur], dt * 16 + li, ks * 64 + lg * 16), dk_acc[dt]);
        }
    }
  }
  // fp32 partials: ws[split][b][hkv][kg][d]
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kg = k0 + lg * 4 + r;
    const int64_t base = ((((int64_t)split * B + b) * Hkv + hkv) * S + kg) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      wsK[base + dt * 16 + li] = dk_acc[dt][r];
      wsV[base + dt * 16 + li] = dv_acc[dt][r];
    }
  }
}

// reduce fp32 split-partials -> bf16 dK/dV in [B,S,Hkv,D]
__global__ void dkv_reduce_kernel(const float* __restrict__ wsK,
                                  const float* __restrict__ wsV,
                                  bf16* __restrict__ dK, bf16* __restrict__ dV,
                                  int B, int Hkv, int S, int D, int splits) {
  const int64_t n = (int64_t)B * Hkv * S * D;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float k = 0.f, v = 0.f;
    for (int sp = 0; sp < splits; ++sp) {
      k += wsK[(int64_t)sp * n + i];
      v += wsV[(int64_t)sp * n + i];
    }
    // i decodes as [b][hkv][s][d] -> output [b][s][hkv][d]
    const int d = (int)(i % D);
    int64_t r = i / D;
    const int sq = (int)(r % S);
    r /= S;
    const int hk = (int)(r % Hkv);
    const int bb = (int)(r / Hkv);
    const int64_t o = (((int64_t)(bb * S + sq)) * Hkv + hk) * D + d;
    dK[o] = f2bf(k);
    dV[o] = f2bf(v);
  }
}

// ------------------------------------------------------------------ host API
#define DISPATCH_D(KER, ...)                                            \
  do {                                                                  \
    if (D == 128)                                                       \
      hipLaunchKernelGGL(KER<128>, dim3(grid), dim3(256), 0, stream,    \
                         __VA_ARGS__);                                  \
    else                                                                \
      hipLaunchKernelGGL(KER<64>, dim3(grid), dim3(256), 0, stream,     \
                         __VA_ARGS__);                                  \
  } while (0)

PRIME_API int prime_flash_fwd(hipStream_t stream, const void* Q, const void* K,
                              const void* Vt, void* O, void* lse, int64_t B,
                              int64_t H, int64_t Hkv, int64_t S, int64_t D,
                              double scale, int64_t causal,
                              int64_t sqb, int64_t sqs, int64_t sqh,
                              int64_t skb, int64_t sks, int64_t skh) {
  if (S % 64 != 0 || (D != 64 && D != 128)) return hipErrorInvalidValue;
  static const char* v5env = getenv("PRIME_ATTN_V5");
  static const char* flagenv = getenv("PRIME_ATTN_FLAGS");
  // measured on MI355X (15-iter sweep): FLAGS=2 417 TF > 3 (406) > 0 (387)
  // > 1 (382) — defer-max pays, setprio slightly negative here. The
  // 3-buffer counted-vmcnt variant (FLAGS=6) measured 401-402 vs 417-418
  // in interleaved A/B: the prefetch-behind-barrier schedule already
  // covers the stage latency, and the third buffer only adds LDS +
  // addressing cost.
  const int flags = flagenv ? atoi(flagenv) : 2;
  const bool use_v5 =
      (D == 128 || D == 64) && (S % 256 == 0) && !(v5env && v5env[0] == '0');
  if (use_v5 && D == 64) {
    const int grid = (int)(B * H * (S / 256));
    switch (flags & 7) {
      case 0:
        hipLaunchKernelGGL((flash_fwd_v5_kernel<64, 0>), dim3(grid),
                           dim3(512), 0, stream, (const bf16*)Q,
                           (const bf16*)K, (const bf16*)Vt, (bf16*)O,
                           (float*)lse, (int)B, (int)H, (int)Hkv, (int)S,
                           (float)scale, (int)causal, sqb, sqs, sqh, skb,
                           sks, skh);
        break;
      default:
        hipLaunchKernelGGL((flash_fwd_v5_kernel<64, 2>), dim3(grid),
                           dim3(512), 0, stream, (const bf16*)Q,
                           (const bf16*)K, (const bf16*)Vt, (bf16*)O,
                           (float*)lse, (int)B, (int)H, (int)Hkv, (int)S,
                           (float)scale, (int)causal, sqb, sqs, sqh, skb,
                           sks, skh);
        break;
    }
    return (int)hipGetLastError();
  }
  if (use_v5) {
    const int grid = (int)(B * H * (S / 256));
#define LAUNCH_V5(F)                                                         \
    hipLaunchKernelGGL((flash_fwd_v5_kernel<128, F>), dim3(grid), dim3(512), \
                       0, stream, (const bf16*)Q, (const bf16*)K,            \
                       (const bf16*)Vt, (bf16*)O, (float*)lse, (int)B,       \
                       (int)H, (int)Hkv, (int)S, (float)scale, (int)causal,  \
                       sqb, sqs, sqh, skb, sks, skh)
    switch (flags & 7) {
      case 0: LAUNCH_V5(0); break;
      case 1: LAUNCH_V5(1); break;
      case 2: LAUNCH_V5(2); break;
      case 3: LAUNCH_V5(3); break;
      case 6: LAUNCH_V5(6); break;  // defer-max + deep prefetch
      case 7: LAUNCH_V5(7); break;
      default: LAUNCH_V5(2); break;
    }
    return (int)hipGetLastError();
  }
#define LAUNCH_FWD(DD, NW)                                                  \
  hipLaunchKernelGGL((flash_fwd_kernel<DD, NW>), dim3(grid),                \
                     dim3(NW * 64), 0, stream, (const bf16*)Q,              \
                     (const bf16*)K, (const bf16*)Vt, (bf16*)O, (float*)lse,\
                     (int)B, (int)H, (int)Hkv, (int)S, (float)scale,        \
                     (int)causal, sqb, sqs, sqh, skb, sks, skh)
  // measured: 8-wave 128-row tiles were SLOWER (190 vs 205 TF/s —
  // redundant K/V fetches were already L2-absorbed; wider barriers and
  // +3% causal waste cost more than the halved staging saved)
  {
    const int grid = (int)(B * H * (S / 64));
    if (D == 128) LAUNCH_FWD(128, 4); else LAUNCH_FWD(64, 4);
  }
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
                                 const void* K, const void* V, const void* Kt,
                                 const void* dO, const void* lse,
                                 const void* delta, void* dQ, int64_t B,
                                 int64_t H, int64_t Hkv, int64_t S, int64_t D,
                                 double scale, int64_t causal,
                                 int64_t sqb, int64_t sqs, int64_t sqh,
                                 int64_t skb, int64_t sks, int64_t skh,
                                 int64_t svb, int64_t svs, int64_t svh) {
  if (S % 64 != 0 || (D != 64 && D != 128)) return hipErrorInvalidValue;
  static const char* v5env = getenv("PRIME_ATTN_V5");
  static const char* flagenv = getenv("PRIME_ATTN_FLAGS");
  const int flags = flagenv ? atoi(flagenv) : 2;
  if ((D == 64) && (S % 256 == 0) && !(v5env && v5env[0] == '0')) {
    const int grid = (int)(B * H * (S / 256));
    hipLaunchKernelGGL((flash_bwd_dq_v5_kernel<64, 0>), dim3(grid),
                       dim3(512), 0, stream, (const bf16*)Q, (const bf16*)K,
                       (const bf16*)V, (const bf16*)Kt, (const bf16*)dO,
                       (const float*)lse, (const float*)delta, (bf16*)dQ,
                       (int)B, (int)H, (int)Hkv, (int)S, (float)scale,
                       (int)causal, sqb, sqs, sqh, skb, sks, skh,
                       svb, svs, svh);
    return (int)hipGetLastError();
  }
  if ((D == 128) && (S % 256 == 0) && !(v5env && v5env[0] == '0')) {
    const int grid = (int)(B * H * (S / 256));
#define LAUNCH_DQ5(F)                                                       \
    hipLaunchKernelGGL((flash_bwd_dq_v5_kernel<128, F>), dim3(grid),        \
                       dim3(512), 0, stream, (const bf16*)Q,                \
                       (const bf16*)K, (const bf16*)V, (const bf16*)Kt,     \
                       (const bf16*)dO, (const float*)lse,                  \
                       (const float*)delta, (bf16*)dQ, (int)B, (int)H,      \
                       (int)Hkv, (int)S, (float)scale, (int)causal,         \
                       sqb, sqs, sqh, skb, sks, skh, svb, svs, svh)
    if (flags & 1) LAUNCH_DQ5(1); else LAUNCH_DQ5(0);
    return (int)hipGetLastError();
  }
  const int grid = (int)(B * H * (S / 64));
  DISPATCH_D(flash_bwd_dq_kernel, (const bf16*)Q, (const bf16*)K,
             (const bf16*)V, (const bf16*)Kt, (const bf16*)dO,
             (const float*)lse, (const float*)delta, (bf16*)dQ, (int)B, (int)H,
             (int)Hkv, (int)S, (float)scale, (int)causal, sqb, sqs, sqh, skb,
             sks, skh, svb, svs, svh);
  return (int)hipGetLastError();
}

PRIME_API int prime_flash_bwd_dkv(hipStream_t stream, const void* Q,
                                  const void* Qt, const void* K, const void* V,
                                  const void* dO, const void* dOt,
                                  const void* lse, const void* delta, void* dK,
                                  void* dV, void* wsK, void* wsV,
                                  int64_t splits, int64_t B, int64_t H,
                                  int64_t Hkv, int64_t S, int64_t D,
                                  double scale, int64_t causal,
                                  int64_t sqb, int64_t sqs, int64_t sqh,
                                  int64_t skb, int64_t sks, int64_t skh,
                                  int64_t svb, int64_t svs, int64_t svh) {
  if (S % 64 != 0 || (D != 64 && D != 128)) return hipErrorInvalidValue;
  // 8-wave blocks: one q/do/qt/dot stage feeds 128 k-rows (half the
  // staging traffic of the 4-wave version); fall back to 4 waves when the
  // sequence doesn't tile by 128
  // 32x32 swapped dkv measured SLOWER than the 16x16 split-q kernel
  // (160 vs 228 TF: the loop-invariant K/V row fragments don't fit the
  // register budget, and demoting them to inline L2 loads puts ~16 global
  // loads in every q-step's MFMA dependency chain). Opt-in for further
  // work via PRIME_ATTN_DKV5=1; see profiles/10b_1gpu_profile.md.
  static const char* v5denv = getenv("PRIME_ATTN_DKV5");
  static const char* fenv = getenv("PRIME_ATTN_FLAGS");
  const int flags = fenv ? atoi(fenv) : 2;
  if ((D == 128) && (S % 256 == 0) && (v5denv && v5denv[0] == '1')) {
    const int sp = (int)(splits > 4 ? 4 : splits);
    const int grid = (int)(B * Hkv * (S / 256) * sp);
#define LAUNCH_DKV5(F)                                                       \
    hipLaunchKernelGGL((flash_bwd_dkv_v5_kernel<128, F>), dim3(grid),        \
                       dim3(512), 0, stream, (const bf16*)Q,                 \
                       (const bf16*)Qt, (const bf16*)K, (const bf16*)V,      \
                       (const bf16*)dO, (const bf16*)dOt,                    \
                       (const float*)lse, (const float*)delta, (float*)wsK,  \
                       (float*)wsV, (int)B, (int)H, (int)Hkv, (int)S,        \
                       (float)scale, (int)causal, sp, sqb, sqs, sqh,         \
                       skb, sks, skh, svb, svs, svh)
    if (flags & 1) LAUNCH_DKV5(1); else LAUNCH_DKV5(0);
    int err5 = hipGetLastError();
    if (err5) return err5;
    int rgrid5 = prime_grid(B * Hkv * S * D, 256);
    hipLaunchKernelGGL(dkv_reduce_kernel, dim3(rgrid5), dim3(256), 0, stream,
                       (const float*)wsK, (const float*)wsV, (bf16*)dK,
                       (bf16*)dV, (int)B, (int)Hkv, (int)S, (int)D, sp);
    return (int)hipGetLastError();
  }
  static const char* nw8env = getenv("PRIME_ATTN_DKV8");
  const bool nw8 = (S % 128 == 0) && !(nw8env && nw8env[0] == '0');
#define LAUNCH_DKV(DD, NWV, DB)                                              \
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<DD, NWV, DB>),                  \
                       dim3((int)(B * Hkv * (S / (NWV * 16)) * splits)),     \
                       dim3(NWV * 64), 0, stream, (const bf16*)Q,            \
                       (const bf16*)Qt, (const bf16*)K, (const bf16*)V,      \
                       (const bf16*)dO, (const bf16*)dOt,                    \
                       (const float*)lse, (const float*)delta, (float*)wsK,  \
                       (float*)wsV, (int)B, (int)H, (int)Hkv, (int)S,        \
                       (float)scale, (int)causal, (int)splits, sqb, sqs,     \
                       sqh, skb, sks, skh, svb, svs, svh)
  static const char* dbenv = getenv("PRIME_ATTN_DKV_DBUF");
  const bool dbuf = !(dbenv && dbenv[0] == '0');
  if (nw8 && dbuf) {
    if (D == 128) LAUNCH_DKV(128, 8, 2); else LAUNCH_DKV(64, 8, 2);
  } else if (nw8) {
    if (D == 128) LAUNCH_DKV(128, 8, 1); else LAUNCH_DKV(64, 8, 1);
  } else {
    if (D == 128) LAUNCH_DKV(128, 4, 1); else LAUNCH_DKV(64, 4, 1);
  }
  int err = hipGetLastError();
  if (err) return err;
  int rgrid = prime_grid(B * Hkv * S * D, 256);
  hipLaunchKernelGGL(dkv_reduce_kernel, dim3(rgrid), dim3(256), 0, stream,
                     (const float*)wsK, (const float*)wsV, (bf16*)dK,
                     (bf16*)dV, (int)B, (int)Hkv, (int)S, (int)D, (int)splits);
  return (int)hipGetLastError();
}
