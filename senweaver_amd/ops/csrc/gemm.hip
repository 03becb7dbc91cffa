// bf16 MFMA GEMM for gfx950 (MI355X): C[M,N] = A[M,K] @ B[N,K]^T
//
// This is the backbone projection GEMM (Llama weights are stored
// [out_features, in_features], so both operands are K-contiguous — the
// "B^T input" form whose fragments are plain 16-byte row loads).
//
// Structure (the CDNA4 guide's "step-3" LDS-staged design):
//   - 128x128 output tile per 256-thread block (4 waves, 2x2; each wave owns
//     a 64x64 quadrant = 4x4 fragments of v_mfma_f32_16x16x32_bf16)
//   - K-loop in BK=64 steps; A-tile and B-tile (16 KiB each) staged by
//     global_load_lds_dwordx4 (direct HBM->LDS DMA, 16 B/lane), double
//     buffered; one __syncthreads per K-step (its vmcnt(0) drains the DMA)
//   - LDS XOR swizzle (chunk ^= row&7) applied on the *source* address and
//     the ds_read address (glds writes lane-linear, so the swizzle cannot go
//     on the LDS destination) — keeps ds_read_b128 bank conflicts <=2-way
//   - XCD-aware bijective blockIdx remap so neighboring tiles (sharing A/B
//     panels) land on the same XCD's L2 (8 XCDs with private 4 MiB L2s)
//
// f32 accumulation in AGPRs; bf16 output.  M,N % 128 == 0, K % 64 == 0
// (host wrapper pads M; Llama-3 dims satisfy N,K natively).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM 128
#define BN 128
#define BK 64
#define WARPS 4

// chunk = 16 bytes = 8 bf16.  A row of BK=64 bf16 is 8 chunks.
#define ROW_CHUNKS 8

__device__ __forceinline__ void stage_tile_glds(
    const ushort* __restrict__ src,  // tile origin: &X[row0*K + k0]
    long long ldK,                   // leading dimension (elements)
    ushort* lds_tile,                // 128*64 bf16, lane-linear destination
    int tid) {
  // 128 rows x 8 chunks = 1024 chunks; 256 threads -> 4 glds each.
  // Linear slot s holds global chunk (row = s/8, c = (s%8) ^ (row&7)):
  // inverse-swizzled source + swizzled read = same involution (rule 21).
  // glds lane destination = wave-uniform base + lane*16, so the base must
  // include this wave's 64-chunk sub-block, not just the iteration offset.
  const int wave_chunk = tid & ~63;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int s = i * 256 + tid;
    const int row = s >> 3;
    const int c = (s & 7) ^ (row & 7);
    const ushort* g = src + (long long)row * ldK + c * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_tile + (long long)(i * 256 + wave_chunk) * 8),
        16, 0, 0);
  }
}

// ds_read_b128 of logical (row, chunk c) from a swizzled tile
__device__ __forceinline__ short8 read_frag(const ushort* lds_tile, int row, int c) {
  const int slot = row * ROW_CHUNKS + (c ^ (row & 7));
  return *reinterpret_cast<const short8*>(lds_tile + slot * 8);
}

extern "C" __global__ void __launch_bounds__(256, 2)
gemm_bt_bf16_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                    ushort* __restrict__ C, int M, int N, int K) {
  // ---- XCD-aware bijective remap of the flat workgroup id (T1) ----
  const int nwg = (M / BM) * (N / BN);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / BN;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int wid = tid >> 6;      // wave 0..3
  const int lane = tid & 63;
  const int wm = wid >> 1;       // 2x2 wave grid: quadrant row
  const int wn = wid & 1;        // quadrant col

  __shared__ __attribute__((aligned(16))) ushort lds[2][2][BM * BK];  // [buf][A/B]

  const ushort* Atile = A + (long long)tile_m * BM * K;
  const ushort* Btile = B + (long long)tile_n * BN * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int m_base = wm * 64;  // within tile
  const int n_base = wn * 64;
  const int frag_row = lane & 15;      // m or n within 16
  const int frag_kgrp = lane >> 4;     // k-group 0..3 (8 bf16 each)

  const int ntiles = K / BK;
  // Prologue: stage tile 0 into buffer 0
  stage_tile_glds(Atile, K, lds[0][0], tid);
  stage_tile_glds(Btile, K, lds[0][1], tid);
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage_tile_glds(Atile + (long long)(t + 1) * BK, K, lds[buf ^ 1][0], tid);
      stage_tile_glds(Btile + (long long)(t + 1) * BK, K, lds[buf ^ 1][1], tid);
    }
    const ushort* Al = lds[buf][0];
    const ushort* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two K=32 MFMA steps per BK=64
      short8 af[4], bf[4];
      const int c = kk * 4 + frag_kgrp;  // logical chunk of this lane's 8 bf16
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = read_frag(Al, m_base + mi * 16 + frag_row, c);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = read_frag(Bl, n_base + ni * 16 + frag_row, c);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // drains the in-flight glds (vmcnt 0) + barrier
    buf ^= 1;
  }

  // ---- Epilogue: C[m][n], C-fragment map row=(lane>>4)*4+e, col=lane&15 ----
  const long long c_row0 = (long long)tile_m * BM + m_base + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * BN + n_base + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const long long row = c_row0 + mi * 16 + e;
      ushort* crow = C + row * N;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        crow[c_col0 + ni * 16] = f2bf(acc[mi][ni][e]);
    }
  }
}

// ---------------------------------------------------------------------------
// 256x256-tile variant — the "BK=64 top tier" structure from the CDNA4
// guide: 8 waves (2Mx4N, each owning 128x64 = 8x4 fragments), 128 KiB LDS
// (1 block/CU), glds double-buffer, ONE __syncthreads per K-step.  Measured
// class: ~1.15-1.22 PF/s bf16 at 4-8k shapes (vs ~0.9 for the 128^2 tile).
// ---------------------------------------------------------------------------
__device__ __forceinline__ void stage_tile_glds_512(
    const ushort* __restrict__ src, long long ldK, ushort* lds_tile, int tid) {
  // 256 rows x 8 chunks = 2048 chunks staged by 512 threads -> 4 glds each
  const int wave_chunk = tid & ~63;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int s = i * 512 + tid;
    const int row = s >> 3;
    const int c = (s & 7) ^ (row & 7);
    const ushort* g = src + (long long)row * ldK + c * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_tile + (long long)(i * 512 + wave_chunk) * 8),
        16, 0, 0);
  }
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_256_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                        ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / 256) * (N / 256);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / 256;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;  // 0..1
  const int wn = wid & 3;   // 0..3

  __shared__ __attribute__((aligned(16))) ushort lds[2][2][256 * 64];

  const ushort* Atile = A + (long long)tile_m * 256 * K;
  const ushort* Btile = B + (long long)tile_n * 256 * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int m_base = wm * 128;
  const int n_base = wn * 64;
  const int frag_row = lane & 15;
  const int frag_kgrp = lane >> 4;

  const int ntiles = K / BK;
  stage_tile_glds_512(Atile, K, lds[0][0], tid);
  stage_tile_glds_512(Btile, K, lds[0][1], tid);
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage_tile_glds_512(Atile + (long long)(t + 1) * BK, K, lds[buf ^ 1][0], tid);
      stage_tile_glds_512(Btile + (long long)(t + 1) * BK, K, lds[buf ^ 1][1], tid);
    }
    const ushort* Al = lds[buf][0];
    const ushort* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      short8 af[8], bf[4];
      const int c = kk * 4 + frag_kgrp;
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
        af[mi] = read_frag(Al, m_base + mi * 16 + frag_row, c);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = read_frag(Bl, n_base + ni * 16 + frag_row, c);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    buf ^= 1;
  }

  const long long c_row0 = (long long)tile_m * 256 + m_base + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * 256 + n_base + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const long long row = c_row0 + mi * 16 + e;
      ushort* crow = C + row * N;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        crow[c_col0 + ni * 16] = f2bf(acc[mi][ni][e]);
    }
  }
}

// ---------------------------------------------------------------------------
// Decode GEMV: M <= 16 (beam decode).  One WAVE per output row n, lanes
// stride K (coalesced 16 B/lane row reads — the previous row-per-lane form
// was ~4x off the bandwidth bound), f32 accumulation, one wave reduction
// per (m, n) at the end.  N/4 blocks fill the chip without split-K, so the
// partial-reduce pass is gone too.  A (M x K, <=128 KiB) streams from L2.
// ---------------------------------------------------------------------------
// concrete instantiations with C linkage for the bindings TU
#define GEMV2_INST(MM)                                                        \
  extern "C" __global__ void __launch_bounds__(256)                           \
  gemv_bt_bf16_v2_m##MM(const ushort* A, const ushort* B, ushort* C, int M,   \
                        int N, int K) {                                       \
    const int wid = threadIdx.x >> 6;                                         \
    const int lane = threadIdx.x & 63;                                        \
    const int n = blockIdx.x * 4 + wid;                                       \
    if (n >= N) return;                                                       \
    const ushort* brow = B + (long long)n * K;                                \
    float acc[MM];                                                            \
    _Pragma("unroll") for (int m = 0; m < MM; ++m) acc[m] = 0.f;              \
    /* 2-deep software pipeline: the runtime-K loop otherwise keeps ONE     \
       weight load in flight per lane (latency-bound on the small rows) */   \
    short8 cur_ = __builtin_nontemporal_load(                                 \
        reinterpret_cast<const short8*>(brow + lane * 8));                    \
    for (int k = lane * 8; k < K; k += 64 * 8) {                              \
      short8 nxt_;                                                            \
      if (k + 64 * 8 < K)                                                     \
        nxt_ = __builtin_nontemporal_load(                                    \
            reinterpret_cast<const short8*>(brow + k + 64 * 8));              \
      bf16x8 bv = *reinterpret_cast<const bf16x8*>(&cur_);                    \
      float bfv[8];                                                           \
      _Pragma("unroll") for (int j = 0; j < 8; ++j) bfv[j] = bf2f(bv.v[j]);   \
      _Pragma("unroll") for (int m = 0; m < MM; ++m) {                        \
        bf16x8 av = *reinterpret_cast<const bf16x8*>(A + (long long)m * K + k); \
        float s = 0.f;                                                        \
        _Pragma("unroll") for (int j = 0; j < 8; ++j)                         \
            s += bf2f(av.v[j]) * bfv[j];                                      \
        acc[m] += s;                                                          \
      }                                                                       \
      cur_ = nxt_;                                                            \
    }                                                                         \
    _Pragma("unroll") for (int m = 0; m < MM; ++m) {                          \
      float v = wave_reduce_sum(acc[m]);                                      \
      if (lane == 0 && m < M) C[(long long)m * N + n] = f2bf(v);              \
    }                                                                         \
  }

// ---------------------------------------------------------------------------
// Decode GEMV v3 — loader/consumer LDS-DMA streaming engine (M = 1).
// The v2 form (one wave per row, 2-deep nt register pipeline) measures ~85%
// of the achievable HBM rate: its in-flight requests are single 1 KiB row
// chunks scattered across many DRAM pages.  The microarch price list
// (ldsdma-fill / nt-weights rows) measures 6.5-6.8 TB/s chip-wide when each
// CU streams CONTIGUOUS 16 KiB fills via global_load_lds with nt.  Here a
// block owns 16 consecutive output rows and walks K in 512-element chunks:
// slot = 16 rows x 1 KiB, ring of 5 slots in LDS (<= 3 in flight keeps the
// counted vmcnt under the 6-bit 63 cap), wave 3 is loader AND consumer, all
// 4 waves split the 16 rows (16 lanes per row, 32 elems per lane per slot),
// x is staged to LDS once up front.  One barrier per slot; counted
// s_waitcnt vmcnt(16*inflight) before it (wait-then-barrier).
// Constraints: M == 1, K % 512 == 0, N % 16 == 0.
// ---------------------------------------------------------------------------
#define GV3_KW 512         /* chunk elems per row per slot */

template <int XCAP, int RD>
static __device__ __forceinline__ void gemv3_body(
    const ushort* __restrict__ A, const ushort* __restrict__ B,
    ushort* __restrict__ C, int N, int K) {
  // 8 rows per block, slot = 8 rows x 1 KiB = 8 KiB, ring of RD slots.
  // K <= 4096: ring 7*8 + x 8 = 64 KiB -> TWO blocks/CU (one block's
  // barrier/latency stalls hide under the other's stream).  K <= 14336
  // (down-proj): x is 28 KiB -> one block/CU.
  const int n0 = blockIdx.x * 8;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;

  __shared__ __attribute__((aligned(16))) ushort gv3_smem[RD * 8 * GV3_KW + XCAP];
  ushort* slots = gv3_smem;
  ushort* x_lds = gv3_smem + RD * 8 * GV3_KW;

  const int nslots = K / GV3_KW;

  if (wid < 3) {
    // stage x (1 KiB chunks, lane-linear)
    for (int c = wid; c < nslots; c += 3)
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)
              (A + (long long)c * GV3_KW + lane * 8),
          (__attribute__((address_space(3))) unsigned int*)
              (x_lds + (long long)c * GV3_KW + lane * 8),
          16, 0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  } else {
    // loader prologue: slots 0..RD-2 (8 nt glds each), then slot 0 landed
    for (int t = 0; t < RD - 1 && t < nslots; ++t) {
      const ushort* src0 = B + (long long)n0 * K + (long long)t * GV3_KW;
      ushort* dst0 = slots + (t % RD) * 8 * GV3_KW;
#pragma unroll
      for (int r = 0; r < 8; ++r)
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)
                (src0 + (long long)r * K + lane * 8),
            (__attribute__((address_space(3))) unsigned int*)
                (dst0 + r * GV3_KW + lane * 8),
            16, 0, 2 /* nt */);
    }
    const int inflight0 = ((RD - 2) < (nslots - 1) ? (RD - 2) : (nslots - 1));
    switch (inflight0 < 0 ? 0 : inflight0) {
      case 0: asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); break;
      case 1: asm volatile("s_waitcnt vmcnt(8)" ::: "memory"); break;
      case 2: asm volatile("s_waitcnt vmcnt(16)" ::: "memory"); break;
      case 3: asm volatile("s_waitcnt vmcnt(24)" ::: "memory"); break;
      case 4: asm volatile("s_waitcnt vmcnt(32)" ::: "memory"); break;
      default: asm volatile("s_waitcnt vmcnt(40)" ::: "memory"); break;
    }
  }
  __builtin_amdgcn_s_barrier();

  // consumer: wave w owns rows 2w, 2w+1; reads are lane-linear b128
  // (conflict-free — the j*32 strided form measured a 16-way bank storm)
  float acc0 = 0.f, acc1 = 0.f;
  const int r0 = wid * 2;
  for (int t = 0; t < nslots; ++t) {
    const ushort* slot = slots + (t % RD) * 8 * GV3_KW;
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(
        x_lds + (long long)t * GV3_KW + lane * 8);
    bf16x8 w0 = *reinterpret_cast<const bf16x8*>(slot + r0 * GV3_KW + lane * 8);
    bf16x8 w1 = *reinterpret_cast<const bf16x8*>(slot + (r0 + 1) * GV3_KW + lane * 8);
    float xf[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) xf[e] = bf2f(xv.v[e]);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc0 += bf2f(w0.v[e]) * xf[e];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc1 += bf2f(w1.v[e]) * xf[e];
    if (wid == 3) {
      const int nxt = t + RD - 1;
      if (nxt < nslots) {
        const ushort* src0 = B + (long long)n0 * K + (long long)nxt * GV3_KW;
        ushort* dst0 = slots + (nxt % RD) * 8 * GV3_KW;
#pragma unroll
        for (int r = 0; r < 8; ++r)
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) unsigned int*)
                  (src0 + (long long)r * K + lane * 8),
              (__attribute__((address_space(3))) unsigned int*)
                  (dst0 + r * GV3_KW + lane * 8),
              16, 0, 2);
      }
      const int last = (nxt < nslots ? nxt : nslots - 1);
      const int inflight = last - (t + 1);
      switch (inflight < 0 ? 0 : inflight) {
        case 0: asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); break;
        case 1: asm volatile("s_waitcnt vmcnt(8)" ::: "memory"); break;
        case 2: asm volatile("s_waitcnt vmcnt(16)" ::: "memory"); break;
        case 3: asm volatile("s_waitcnt vmcnt(24)" ::: "memory"); break;
        case 4: asm volatile("s_waitcnt vmcnt(32)" ::: "memory"); break;
        default: asm volatile("s_waitcnt vmcnt(40)" ::: "memory"); break;
      }
    }
    asm volatile("" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }

  acc0 = wave_reduce_sum(acc0);
  acc1 = wave_reduce_sum(acc1);
  if (lane == 0 && n0 + r0 < N) C[n0 + r0] = f2bf(acc0);
  if (lane == 0 && n0 + r0 + 1 < N) C[n0 + r0 + 1] = f2bf(acc1);
}

// two slots per barrier round: the 1-slot cadence measured ~0.7 us/slot
// against a 0.32 us fill (barrier+loader-wait dominated); pairing slots
// halves the sync cost per byte.  Ring 6 (steps in flight: consuming,
// landed, in-flight = 3 steps x 2 slots).  Needs nslots even (K % 1024).
template <int XCAP, int FUSE_SWIGLU = 0>
static __device__ __forceinline__ void gemv3_body2(
    const ushort* __restrict__ A, const ushort* __restrict__ B,
    ushort* __restrict__ C, int N, int K) {
  const int n0 = blockIdx.x * 8;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;

  __shared__ __attribute__((aligned(16))) ushort gv3_smem[6 * 8 * GV3_KW + XCAP];
  ushort* slots = gv3_smem;
  ushort* x_lds = gv3_smem + 6 * 8 * GV3_KW;

  const int nsteps = K / (2 * GV3_KW);

#define GV3_ISSUE_STEP(ST)                                                    \
  do {                                                                        \
    const ushort* s0_ = B + (long long)n0 * K + (long long)(ST) * 2 * GV3_KW; \
    ushort* d0_ = slots + (((ST) * 2) % 6) * 8 * GV3_KW;                      \
    _Pragma("unroll") for (int r = 0; r < 8; ++r) {                           \
      __builtin_amdgcn_global_load_lds(                                       \
          (const __attribute__((address_space(1))) unsigned int*)             \
              (s0_ + (long long)r * K + lane * 8),                            \
          (__attribute__((address_space(3))) unsigned int*)                   \
              (d0_ + r * GV3_KW + lane * 8),                                  \
          16, 0, 2);                                                          \
      __builtin_amdgcn_global_load_lds(                                       \
          (const __attribute__((address_space(1))) unsigned int*)             \
              (s0_ + (long long)r * K + GV3_KW + lane * 8),                   \
          (__attribute__((address_space(3))) unsigned int*)                   \
              (d0_ + 8 * GV3_KW + r * GV3_KW + lane * 8),                     \
          16, 0, 2);                                                          \
    }                                                                         \
  } while (0)

  if (wid < 3) {
    if (FUSE_SWIGLU) {
      // x = silu(g) * u computed on the way in: A = gateup [1, 2K],
      // g = A[0..K), u = A[K..2K) — replaces the standalone swiglu
      // kernel (one ~5 us launch per decode layer)
      for (int c = wid; c < K / GV3_KW; c += 3) {
        bf16x8 g = *reinterpret_cast<const bf16x8*>(
            A + (long long)c * GV3_KW + lane * 8);
        bf16x8 u = *reinterpret_cast<const bf16x8*>(
            A + (long long)K + (long long)c * GV3_KW + lane * 8);
        bf16x8 out;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float gf = bf2f(g.v[e]);
          out.v[e] = f2bf(gf / (1.f + __expf(-gf)) * bf2f(u.v[e]));
        }
        *reinterpret_cast<bf16x8*>(x_lds + (long long)c * GV3_KW + lane * 8) =
            out;
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // writes visible
    } else {
      for (int c = wid; c < K / GV3_KW; c += 3)
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)
                (A + (long long)c * GV3_KW + lane * 8),
            (__attribute__((address_space(3))) unsigned int*)
                (x_lds + (long long)c * GV3_KW + lane * 8),
            16, 0, 0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
  } else {
    GV3_ISSUE_STEP(0);
    if (1 < nsteps) {
      GV3_ISSUE_STEP(1);
      asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
  }
  __builtin_amdgcn_s_barrier();

  float acc0 = 0.f, acc1 = 0.f;
  const int r0 = wid * 2;
  for (int st = 0; st < nsteps; ++st) {
    const ushort* slot0 = slots + ((st * 2) % 6) * 8 * GV3_KW;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const ushort* slot = slot0 + h * 8 * GV3_KW;
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(
          x_lds + ((long long)st * 2 + h) * GV3_KW + lane * 8);
      bf16x8 w0 = *reinterpret_cast<const bf16x8*>(slot + r0 * GV3_KW + lane * 8);
      bf16x8 w1 = *reinterpret_cast<const bf16x8*>(slot + (r0 + 1) * GV3_KW + lane * 8);
      float xf[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) xf[e] = bf2f(xv.v[e]);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc0 += bf2f(w0.v[e]) * xf[e];
#pragma unroll
      for (int e = 0; e < 8; ++e) acc1 += bf2f(w1.v[e]) * xf[e];
    }
    if (wid == 3) {
      const int nxt = st + 2;
      if (nxt < nsteps) {
        GV3_ISSUE_STEP(nxt);
        // outstanding: step st+1 (16, oldest) + step st+2 (16)
        asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
      } else if (st + 1 < nsteps) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
    }
    asm volatile("" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
#undef GV3_ISSUE_STEP

  acc0 = wave_reduce_sum(acc0);
  acc1 = wave_reduce_sum(acc1);
  if (lane == 0 && n0 + r0 < N) C[n0 + r0] = f2bf(acc0);
  if (lane == 0 && n0 + r0 + 1 < N) C[n0 + r0 + 1] = f2bf(acc1);
}

extern "C" __global__ void __launch_bounds__(256)
gemv_bt_bf16_v3_m1(const ushort* __restrict__ A, const ushort* __restrict__ B,
                   ushort* __restrict__ C, int M, int N, int K) {
  // 56 KiB LDS -> two blocks/CU; two slots per barrier round
  gemv3_body2<4096>(A, B, C, N, K);
}

extern "C" __global__ void __launch_bounds__(256)
gemv_bt_bf16_v3w_m1(const ushort* __restrict__ A, const ushort* __restrict__ B,
                    ushort* __restrict__ C, int M, int N, int K) {
  // wide-K (down-proj): 76 KiB LDS -> one block/CU
  gemv3_body2<14336>(A, B, C, N, K);
}

extern "C" __global__ void __launch_bounds__(256)
swiglu_gemv_bt_bf16_m1(const ushort* __restrict__ GU,
                       const ushort* __restrict__ B,
                       ushort* __restrict__ C, int M, int N, int K) {
  // fused silu(g)*u + down-proj GEMV (decode): GU = gateup [1, 2K]
  gemv3_body2<14336, 1>(GU, B, C, N, K);
}

// ---------------------------------------------------------------------------
// fp8-weight GEMV (decode, M = 1): W rows stored OCP e4m3 with a per-row
// f32 scale (the model's existing fp8 weight format); x stays bf16 — no
// per-step activation quant kernel and no M-pad through the 128-tile fp8
// GEMM (which measured 101 tok/s vs 267 bf16 in decode).  Weight bytes
// halve, and decode GEMV is weight-bandwidth-bound.
// One wave per output row; 16 fp8 bytes per lane per iter via
// v_cvt_pk_f32_fp8 pairs; K % 1024 == 0.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float f32x2_t;
typedef __attribute__((ext_vector_type(4))) unsigned uint4_t;

// TWO rows per wave: fp8 rows are half the bytes of bf16 rows, so a
// K=4096 row is only 4 pipeline iterations — latency-bound with one
// row's 2-deep pipeline.  Two interleaved rows double the in-flight
// loads at the same depth.
template <int MM, int R>  // R rows per wave (R=2 for batched: doubles
                           // in-flight loads on the short fp8 rows; R=1
                           // keeps the M=1 path's measured 332 tok/s)
static __device__ __forceinline__ void gemv_fp8w_body(
    const ushort* __restrict__ X, const unsigned char* __restrict__ Bq,
    const float* __restrict__ Bs, ushort* __restrict__ C, int M, int N,
    int K) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = (blockIdx.x * 4 + wid) * R;
  if (n >= N) return;
  const unsigned char* brow0 = Bq + (long long)n * K;
  const unsigned char* brow1 = brow0 + (R > 1 ? K : 0);
  float acc[MM][R];
#pragma unroll
  for (int m = 0; m < MM; ++m)
#pragma unroll
    for (int r = 0; r < R; ++r) acc[m][r] = 0.f;
  uint4_t cur0 = __builtin_nontemporal_load(
      reinterpret_cast<const uint4_t*>(brow0 + lane * 16));
  uint4_t cur1 = {};
  if (R > 1)
    cur1 = __builtin_nontemporal_load(
        reinterpret_cast<const uint4_t*>(brow1 + lane * 16));
  for (int k = lane * 16; k < K; k += 64 * 16) {
    uint4_t nxt0, nxt1;
    if (k + 64 * 16 < K) {
      nxt0 = __builtin_nontemporal_load(
          reinterpret_cast<const uint4_t*>(brow0 + k + 64 * 16));
      if (R > 1)
        nxt1 = __builtin_nontemporal_load(
            reinterpret_cast<const uint4_t*>(brow1 + k + 64 * 16));
    }
    float wf0[16], wf1[16];
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      f32x2_t lo0 = __builtin_amdgcn_cvt_pk_f32_fp8(cur0[d], false);
      f32x2_t hi0 = __builtin_amdgcn_cvt_pk_f32_fp8(cur0[d], true);
      wf0[d * 4 + 0] = lo0[0]; wf0[d * 4 + 1] = lo0[1];
      wf0[d * 4 + 2] = hi0[0]; wf0[d * 4 + 3] = hi0[1];
      if (R > 1) {
        f32x2_t lo1 = __builtin_amdgcn_cvt_pk_f32_fp8(cur1[d], false);
        f32x2_t hi1 = __builtin_amdgcn_cvt_pk_f32_fp8(cur1[d], true);
        wf1[d * 4 + 0] = lo1[0]; wf1[d * 4 + 1] = lo1[1];
        wf1[d * 4 + 2] = hi1[0]; wf1[d * 4 + 3] = hi1[1];
      }
    }
#pragma unroll
    for (int m = 0; m < MM; ++m) {
      bf16x8 x0 = *reinterpret_cast<const bf16x8*>(X + (long long)m * K + k);
      bf16x8 x1 = *reinterpret_cast<const bf16x8*>(X + (long long)m * K + k + 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float xf = bf2f(x0.v[e]);
        acc[m][0] += wf0[e] * xf;
        if (R > 1) acc[m][1] += wf1[e] * xf;
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float xf = bf2f(x1.v[e]);
        acc[m][0] += wf0[8 + e] * xf;
        if (R > 1) acc[m][1] += wf1[8 + e] * xf;
      }
    }
    cur0 = nxt0;
    if (R > 1) cur1 = nxt1;
  }
#pragma unroll
  for (int m = 0; m < MM; ++m) {
#pragma unroll
    for (int r = 0; r < R; ++r) {
      const float v = wave_reduce_sum(acc[m][r]);
      if (lane == 0 && m < M && n + r < N)
        C[(long long)m * N + n + r] = f2bf(v * Bs[n + r]);
    }
  }
}

// mxfp8-weight variant: e8m0 per-32-block scales.  A lane's 16 elems per
// iter never straddle a 32-block (k % 16 == 0), so ONE scale byte per lane
// per iter: wf *= 2^(s-127) via ldexpf on the per-iter partial: the scale is
// folded per-iter partial sum (s is block-constant).
template <int MM>
static __device__ __forceinline__ void gemv_mxfp8w_body(
    const ushort* __restrict__ X, const unsigned char* __restrict__ Bq,
    const unsigned char* __restrict__ Bs, ushort* __restrict__ C, int M,
    int N, int K) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;
  const unsigned char* brow = Bq + (long long)n * K;
  const unsigned char* srow = Bs + (long long)n * (K / 32);
  float acc[MM];
#pragma unroll
  for (int m = 0; m < MM; ++m) acc[m] = 0.f;
  uint4_t cur = __builtin_nontemporal_load(
      reinterpret_cast<const uint4_t*>(brow + lane * 16));
  for (int k = lane * 16; k < K; k += 64 * 16) {
    uint4_t nxt;
    if (k + 64 * 16 < K)
      nxt = __builtin_nontemporal_load(
          reinterpret_cast<const uint4_t*>(brow + k + 64 * 16));
    const float scale = exp2f((float)srow[k / 32] - 127.f);
    float wf[16];
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      f32x2_t lo = __builtin_amdgcn_cvt_pk_f32_fp8(cur[d], false);
      f32x2_t hi = __builtin_amdgcn_cvt_pk_f32_fp8(cur[d], true);
      wf[d * 4 + 0] = lo[0];
      wf[d * 4 + 1] = lo[1];
      wf[d * 4 + 2] = hi[0];
      wf[d * 4 + 3] = hi[1];
    }
#pragma unroll
    for (int m = 0; m < MM; ++m) {
      bf16x8 x0 = *reinterpret_cast<const bf16x8*>(X + (long long)m * K + k);
      bf16x8 x1 = *reinterpret_cast<const bf16x8*>(X + (long long)m * K + k + 8);
      float part = 0.f;
#pragma unroll
      for (int e = 0; e < 8; ++e) part += wf[e] * bf2f(x0.v[e]);
#pragma unroll
      for (int e = 0; e < 8; ++e) part += wf[8 + e] * bf2f(x1.v[e]);
      acc[m] += part * scale;
    }
    cur = nxt;
  }
#pragma unroll
  for (int m = 0; m < MM; ++m) {
    const float v = wave_reduce_sum(acc[m]);
    if (lane == 0 && m < M) C[(long long)m * N + n] = f2bf(v);
  }
}

#define GEMV_MXFP8W_INST(MM)                                                  \
  extern "C" __global__ void __launch_bounds__(256)                           \
  gemv_bt_mxfp8w_m##MM(const ushort* X, const unsigned char* Bq,              \
                       const unsigned char* Bs, ushort* C, int M, int N,      \
                       int K) {                                               \
    gemv_mxfp8w_body<MM>(X, Bq, Bs, C, M, N, K);                              \
  }

GEMV_MXFP8W_INST(1)
GEMV_MXFP8W_INST(2)
GEMV_MXFP8W_INST(4)
GEMV_MXFP8W_INST(8)

#define GEMV_FP8W_INST(MM)                                                    \
  extern "C" __global__ void __launch_bounds__(256)                           \
  gemv_bt_fp8w_m##MM(const ushort* X, const unsigned char* Bq,                \
                     const float* Bs, ushort* C, int M, int N, int K) {       \
    gemv_fp8w_body<MM, (MM > 1 ? 2 : 1)>(X, Bq, Bs, C, M, N, K);              \
  }

GEMV_FP8W_INST(1)
GEMV_FP8W_INST(2)
GEMV_FP8W_INST(4)
GEMV_FP8W_INST(8)

GEMV2_INST(1)
GEMV2_INST(2)
GEMV2_INST(4)
GEMV2_INST(8)
GEMV2_INST(16)

// ---------------------------------------------------------------------------
// 256x256-tile on v_mfma_f32_32x32x16_bf16: identical FLOPs, LDS traffic and
// geometry to gemm_bt_bf16_256_kernel (8 waves, wave tile 128x64, BK=64,
// glds dbuf, one barrier/K-tile) but HALF the MFMA instruction count (32
// 32x32x16 per K-tile vs 64 16x16x32) — attacking the issue-stall share the
// PMC log attributes to the 16x16 version.  Fragment maps follow the
// probed/validated attention + MX-kernel layouts: A/B lane row = base+l31,
// k = step*16 + lhi*8 + e; C col = lane&31 (N), row = (r&3)+8*(r>>2)+4*lhi.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(16))) float f32x16g;

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_256x32_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / 256) * (N / 256);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / 256;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;  // 0..1
  const int wn = wid & 3;   // 0..3
  const int l31 = lane & 31;
  const int lhi = lane >> 5;

  __shared__ __attribute__((aligned(16))) ushort lds[2][2][256 * 64];

  const ushort* Atile = A + (long long)tile_m * 256 * K;
  const ushort* Btile = B + (long long)tile_n * 256 * K;

  f32x16g acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;

  const int m_base = wm * 128;
  const int n_base = wn * 64;

  const int ntiles = K / BK;
  stage_tile_glds_512(Atile, K, lds[0][0], tid);
  stage_tile_glds_512(Btile, K, lds[0][1], tid);
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage_tile_glds_512(Atile + (long long)(t + 1) * BK, K, lds[buf ^ 1][0], tid);
      stage_tile_glds_512(Btile + (long long)(t + 1) * BK, K, lds[buf ^ 1][1], tid);
    }
    const ushort* Al = lds[buf][0];
    const ushort* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int stp = 0; stp < 4; ++stp) {  // BK=64 in 4 K=16 steps
      short8 af[4], bf[2];
      const int c = stp * 2 + lhi;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = read_frag(Al, m_base + mi * 32 + l31, c);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        bf[ni] = read_frag(Bl, n_base + ni * 32 + l31, c);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    buf ^= 1;
  }

  const long long c_col0 = (long long)tile_n * 256 + n_base + l31;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long long row = (long long)tile_m * 256 + m_base + mi * 32
                            + (r & 3) + 8 * (r >> 2) + 4 * lhi;
      ushort* crow = C + row * N;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        crow[c_col0 + ni * 32] = f2bf(acc[mi][ni][r]);
    }
}

// ---------------------------------------------------------------------------
// A/B experiment: 256-tile 16x16x32 kernel with explicit scheduling groups —
// __builtin_amdgcn_sched_group_barrier interleaves each MFMA with a DS read
// so fragment-load latency hides under matrix ops instead of clustering.
// SCHED_GROUP_BARRIER(mask, size, syncID): mask 0x8 = MFMA, 0x100 = DS read.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_256sg_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                          ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / 256) * (N / 256);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / 256;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;
  const int wn = wid & 3;

  __shared__ __attribute__((aligned(16))) ushort lds[2][2][256 * 64];

  const ushort* Atile = A + (long long)tile_m * 256 * K;
  const ushort* Btile = B + (long long)tile_n * 256 * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int m_base = wm * 128;
  const int n_base = wn * 64;
  const int frag_row = lane & 15;
  const int frag_kgrp = lane >> 4;

  const int ntiles = K / BK;
  stage_tile_glds_512(Atile, K, lds[0][0], tid);
  stage_tile_glds_512(Btile, K, lds[0][1], tid);
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage_tile_glds_512(Atile + (long long)(t + 1) * BK, K, lds[buf ^ 1][0], tid);
      stage_tile_glds_512(Btile + (long long)(t + 1) * BK, K, lds[buf ^ 1][1], tid);
    }
    const ushort* Al = lds[buf][0];
    const ushort* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      short8 af[8], bf[4];
      const int c = kk * 4 + frag_kgrp;
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
        af[mi] = read_frag(Al, m_base + mi * 16 + frag_row, c);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = read_frag(Bl, n_base + ni * 16 + frag_row, c);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      // interleave: 12 DS reads paired through the 32 MFMAs of this step
#pragma unroll
      for (int g = 0; g < 12; ++g) {
        __builtin_amdgcn_sched_group_barrier(0x100, 1, 0);  // 1 DS read
        __builtin_amdgcn_sched_group_barrier(0x008, 2, 0);  // 2 MFMAs
      }
      __builtin_amdgcn_sched_group_barrier(0x008, 8, 0);    // trailing MFMAs
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    buf ^= 1;
  }

  const long long c_row0 = (long long)tile_m * 256 + m_base + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * 256 + n_base + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const long long row = c_row0 + mi * 16 + e;
      ushort* crow = C + row * N;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        crow[c_col0 + ni * 16] = f2bf(acc[mi][ni][e]);
    }
  }
}

// ---------------------------------------------------------------------------
// Decode fused residual-add + RMSNorm + GEMV:
//   t[m] = x[m] (+ res_in[m]);  C[m][n] = (sum_k t*normw*B[n]) * rsqrt(ss/K+eps)
// RMSNorm is a per-row LINEAR scale, so the raw dot and the sum of squares
// accumulate in the SAME k-pass and the scale applies after the wave
// reduction — one kernel replaces (fused_add_rmsnorm + gemv), two of the
// ~5 us launches per decode layer.  Block 0 additionally writes
// res_out[m] = t[m] (a separate buffer: no cross-block write/read race).
// ---------------------------------------------------------------------------
#define GEMV_NORM_INST(MM)                                                    \
  extern "C" __global__ void __launch_bounds__(256)                           \
  gemv_norm_bt_bf16_m##MM(const ushort* x, const ushort* res_in,              \
                          const ushort* normw, const ushort* B, ushort* C,    \
                          ushort* res_out, int M, int N, int K, float eps) {  \
    const int wid = threadIdx.x >> 6;                                         \
    const int lane = threadIdx.x & 63;                                        \
    const int n = blockIdx.x * 4 + wid;                                       \
    if (n >= N) return;                                                       \
    const ushort* brow = B + (long long)n * K;                                \
    float dot[MM], ss[MM];                                                    \
    _Pragma("unroll") for (int m = 0; m < MM; ++m) { dot[m] = 0.f; ss[m] = 0.f; } \
    for (int k = lane * 8; k < K; k += 64 * 8) {                              \
      short8 bv_ = __builtin_nontemporal_load(                                \
          reinterpret_cast<const short8*>(brow + k));                         \
      bf16x8 bv = *reinterpret_cast<const bf16x8*>(&bv_);                     \
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(normw + k);                \
      float bw[8];                                                            \
      _Pragma("unroll") for (int j = 0; j < 8; ++j)                           \
          bw[j] = bf2f(bv.v[j]) * bf2f(wv.v[j]);                              \
      _Pragma("unroll") for (int m = 0; m < MM; ++m) {                        \
        bf16x8 av = *reinterpret_cast<const bf16x8*>(x + (long long)m * K + k); \
        float t[8];                                                           \
        if (res_in != nullptr) {                                              \
          bf16x8 rv = *reinterpret_cast<const bf16x8*>(res_in + (long long)m * K + k); \
          _Pragma("unroll") for (int j = 0; j < 8; ++j)                       \
              t[j] = bf2f(av.v[j]) + bf2f(rv.v[j]);                           \
        } else {                                                              \
          _Pragma("unroll") for (int j = 0; j < 8; ++j) t[j] = bf2f(av.v[j]); \
        }                                                                     \
        float d = 0.f, s2 = 0.f;                                              \
        _Pragma("unroll") for (int j = 0; j < 8; ++j) {                       \
          d += t[j] * bw[j];                                                  \
          s2 += t[j] * t[j];                                                  \
        }                                                                     \
        dot[m] += d;                                                          \
        ss[m] += s2;                                                          \
      }                                                                       \
    }                                                                         \
    _Pragma("unroll") for (int m = 0; m < MM; ++m) {                          \
      float d = wave_reduce_sum(dot[m]);                                      \
      float s2 = wave_reduce_sum(ss[m]);                                      \
      if (lane == 0 && m < M)                                                 \
        C[(long long)m * N + n] = f2bf(d * rsqrtf(s2 / (float)K + eps));      \
    }                                                                         \
    if (blockIdx.x == 0) {                                                    \
      /* waves stride rows; lanes stride k: res_out = t (vectorized) */       \
      for (int m = wid; m < M; m += 4) {                                      \
        for (int k = lane * 8; k < K; k += 64 * 8) {                          \
          bf16x8 av = *reinterpret_cast<const bf16x8*>(x + (long long)m * K + k); \
          bf16x8 o;                                                           \
          if (res_in != nullptr) {                                            \
            bf16x8 rv = *reinterpret_cast<const bf16x8*>(res_in + (long long)m * K + k); \
            _Pragma("unroll") for (int j = 0; j < 8; ++j)                     \
                o.v[j] = f2bf(bf2f(av.v[j]) + bf2f(rv.v[j]));                 \
          } else {                                                            \
            o = av;                                                           \
          }                                                                   \
          *reinterpret_cast<bf16x8*>(res_out + (long long)m * K + k) = o;     \
        }                                                                     \
      }                                                                       \
    }                                                                         \
  }

GEMV_NORM_INST(1)
GEMV_NORM_INST(2)
GEMV_NORM_INST(4)
GEMV_NORM_INST(8)
GEMV_NORM_INST(16)
