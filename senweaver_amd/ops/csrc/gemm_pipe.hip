// 8-phase software-pipelined bf16 MFMA GEMM for gfx950 (MI355X).
//
//   C[M,N] = A[M,K] @ B[N,K]^T   (both operands K-contiguous, bf16, f32 acc)
//
// This is the deep-pipelined 256x256 structure the CDNA4 guide documents as
// the plain-HIP top tier (~1.3-1.5 PF/s on random data), replacing the
// one-barrier-per-K-tile glds double-buffer whose ~1.0-1.2 PF ceiling is the
// single vmcnt(0)+barrier drain per K-step (measured round 1:
// profiles/r01_pmc_summary.txt — 57% MFMA-busy, stall = the barrier drain).
//
// Structure per 512-thread block (8 waves as 2M x 4N, wave tile 128x64):
//   - K advances in BK=64 tiles; each tile's operands are staged as FOUR
//     16 KiB half-tiles (k-split: A/B x k-halves of 32), stream-ordered
//     [B-kh0, A-kh0, B-kh1, A-kh1], double-buffered (2 x 4 slots = 128 KiB).
//   - 4 phases per K-tile, each phase: {4-or-8 ds_read_b128 fragment loads,
//     issue ONE half-tile prefetch (2 global_load_lds_dwordx4/thread),
//     s_barrier, 16 MFMA under s_setprio(1), s_barrier}.  Phases pair one
//     wave's loads with other waves' MFMAs on the same SIMD (role split).
//   - vmcnt is counted, ONCE per K-tile (s_waitcnt vmcnt(6) at the tile
//     boundary = 3 half-tiles left in flight), never 0 in the main loop;
//     raw s_barrier (not __syncthreads) so in-flight DMA crosses barriers.
//     Prefetch runs 3-7 half-tiles ahead; the boundary wait guarantees the
//     whole NEXT tile has landed while tile t+2's halves stream.
//   - Overwrite legality is phase-exact: slot j of the current buffer has
//     its last ds_read at phase j's top, and tile t+2's half j is issued at
//     phase j+1, after the closing barrier of phase j drained every wave's
//     reads (stream order == per-phase consumption order, by construction).
//   - LDS swizzle: a k-half slot is [256 rows][4 chunks of 16 B]; chunk c of
//     row r lands at slot H[c] ^ ((r>>2)&3), H = {0,3,1,2}.  This makes all
//     four of ds_read_b128's 16-lane groups conflict-free for the fragment
//     read pattern (row = base + lane&15, c = lane>>4) — verified per group
//     against the (addr/4)%64 banking of the microarch guide.  The inverse
//     permutation is applied to the global SOURCE address (glds destinations
//     are lane-linear), staying within each row's 64 B k-half segment so
//     cacheline coalescing is preserved.
//   - XCD-aware bijective blockIdx remap (neighbor tiles share L2 panels).
//
// Constraints: M % 256 == 0 (host pads), N % 256 == 0, K % 128 == 0
// (even K-tile count keeps the double-buffer flip compile-time).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define PBM 256
#define PBN 256
#define PBK 64
// one half-tile slot: 256 rows x 32 k = 256 x 4 chunks x 16 B = 16 KiB
#define SLOT_USHORT (256 * 32)
// chunk-permutation tables packed as nibbles: H = {0,3,1,2}, H^-1 = {0,2,3,1}
#define H_PACK 0x2130u
#define HINV_PACK 0x1320u

// Issue one half-tile (16 KiB) as 2 global_load_lds_dwordx4 per thread.
// row/c precomputed per thread outside the loop; k0 = t*64 + kh*32.
__device__ __forceinline__ void pipe_stage_half(
    const ushort* __restrict__ op, long long ldK, int k0,
    ushort* lds_slot, const int row[2], const int cofs[2], int wave_chunk) {
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const ushort* g = op + (long long)row[i] * ldK + k0 + cofs[i];
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_slot +
            (long long)(i * 512 + wave_chunk) * 8),
        16, 0, 0);
  }
}

// Structural variants (within-probe A/B; VAR=0 is the shipped kernel):
//   0: two raw barriers per phase ({reads,glds} | BAR | MFMA | BAR)
//   1: ONE raw barrier per phase ({reads, glds, MFMA} | BAR) — reads and
//      MFMAs co-scheduled inside the segment; overwrite-legality window is
//      still <1 phase because a wave reaches the barrier only after its
//      MFMAs (which drained its reads)
//   2: VAR 0 without s_setprio (isolates T5)
//   3: VAR 1 with static young-half priority (waves 4-7 setprio(1) once)
template <int VAR>
__device__ __forceinline__ void
gemm8ph_body(const ushort* __restrict__ A, const ushort* __restrict__ B,
             ushort* __restrict__ C, int M, int N, int K) {
  // ---- XCD-aware bijective remap (T1) ----
  const int nwg = (M / PBM) * (N / PBN);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / PBN;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;            // 0..1 -> 128-row half of A
  const int wn = wid & 3;             // 0..3 -> 64-row quarter of B
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;         // chunk within a k-half (0..3)

  // slots in stream order per buf: 0=B-kh0, 1=A-kh0, 2=B-kh1, 3=A-kh1
  __shared__ __attribute__((aligned(16))) ushort lds[2][4][SLOT_USHORT];

  const ushort* Atile = A + (long long)tile_m * PBM * K;
  const ushort* Btile = B + (long long)tile_n * PBN * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // fragment-read byte offset within a slot (constant per lane):
  //   addr(row, c) = row*64 + (H[c] ^ ((row>>2)&3))*16, row = base + l15,
  //   and base>>2 == 0 (mod 4) for every frag base, so r4 = (l15>>2)&3.
  const int swz = ((H_PACK >> (kgrp * 4)) & 0xF) ^ ((l15 >> 2) & 3);
  const int frag_off = l15 * 64 + swz * 16;          // bytes
  const int a_off = wm * 128 * 64 + frag_off;        // + mi*16*64
  const int b_off = wn * 64 * 64 + frag_off;         // + ni*16*64

  // staging source coords per thread (i = 0,1): row s>>2, chunk from H^-1
  int st_row[2], st_cofs[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int s = i * 512 + tid;
    const int r = s >> 2;
    const int c = (HINV_PACK >> ((((s & 3) ^ ((r >> 2) & 3)) * 4))) & 0xF;
    st_row[i] = r;
    st_cofs[i] = c * 8;  // elements
  }
  const int wave_chunk = tid & ~63;

  const int ntiles = K / PBK;  // even (K % 128 == 0)

  // ---- prologue: halves 0..3 (tile 0), vmcnt(4); halves 4..6, vmcnt(6) ----
  // half h: tile h>>2, slot h&3; slot&1 ? A : B; k-half (slot>>1)&1
#define ISSUE_HALF(TGT, SLOT, BUF)                                            \
  do {                                                                        \
    if ((TGT) < ntiles) {                                                     \
      const int k0_ = (TGT) * PBK + (((SLOT) >> 1) & 1) * 32;                 \
      if ((SLOT) & 1)                                                         \
        pipe_stage_half(Atile, K, k0_, &lds[(BUF)][(SLOT)][0], st_row,        \
                        st_cofs, wave_chunk);                                 \
      else                                                                    \
        pipe_stage_half(Btile, K, k0_, &lds[(BUF)][(SLOT)][0], st_row,        \
                        st_cofs, wave_chunk);                                 \
    }                                                                         \
  } while (0)

  ISSUE_HALF(0, 0, 0);
  ISSUE_HALF(0, 1, 0);
  ISSUE_HALF(0, 2, 0);
  ISSUE_HALF(0, 3, 0);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  ISSUE_HALF(1, 0, 1);
  ISSUE_HALF(1, 1, 1);
  ISSUE_HALF(1, 2, 1);
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  if (VAR == 3 && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // static young-half priority (T5)

  // ---- main loop: 8 phases per 2 K-tiles, buf flips are compile-time ----
  // Phase p of tile t: ds_read/MFMA quadrant p; issue half (4t + 7 + p).
  //   p0: A-kh0 mi0..3 + B-kh0 (8 reads) | issue A-kh1(t+1) -> buf^1 slot 3
  //   p1: A-kh0 mi4..7      (4 reads)    | issue B-kh0(t+2) -> buf   slot 0
  //   p2: A-kh1 mi0..3 + B-kh1 (8 reads) | issue A-kh0(t+2) -> buf   slot 1
  //   p3: A-kh1 mi4..7      (4 reads)    | issue B-kh1(t+2) -> buf   slot 2
#define LOAD_A4(DST, BUF, KH_SLOT, MI0)                                       \
  _Pragma("unroll") for (int j = 0; j < 4; ++j) {                             \
    DST[j] = *reinterpret_cast<const short8*>(                                \
        reinterpret_cast<const char*>(&lds[(BUF)][(KH_SLOT)][0]) + a_off +    \
        ((MI0) + j) * 1024);                                                  \
  }
#define LOAD_B4(DST, BUF, KH_SLOT)                                            \
  _Pragma("unroll") for (int j = 0; j < 4; ++j) {                             \
    DST[j] = *reinterpret_cast<const short8*>(                                \
        reinterpret_cast<const char*>(&lds[(BUF)][(KH_SLOT)][0]) + b_off +    \
        j * 1024);                                                            \
  }
#define MFMA16(MI0)                                                           \
  if (VAR == 0 || VAR == 2) __builtin_amdgcn_s_barrier();                     \
  if (VAR != 2 && VAR != 3) __builtin_amdgcn_s_setprio(1);                    \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                            \
  _Pragma("unroll") for (int ni = 0; ni < 4; ++ni)                           \
      acc[(MI0) + mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(          \
          af[mi], bf[ni], acc[(MI0) + mi][ni], 0, 0, 0);                      \
  if (VAR != 2 && VAR != 3) __builtin_amdgcn_s_setprio(0);                    \
  __builtin_amdgcn_s_barrier();

#define TILE4(T, BUF)                                                         \
  do {                                                                        \
    short8 af[4], bf[4];                                                      \
    /* phase 0 */                                                             \
    LOAD_A4(af, BUF, 1, 0);                                                   \
    LOAD_B4(bf, BUF, 0);                                                      \
    ISSUE_HALF((T) + 1, 3, (BUF) ^ 1);                                        \
    MFMA16(0);                                                                \
    /* phase 1 */                                                             \
    LOAD_A4(af, BUF, 1, 4);                                                   \
    ISSUE_HALF((T) + 2, 0, BUF);                                              \
    MFMA16(4);                                                                \
    /* phase 2 */                                                             \
    LOAD_A4(af, BUF, 3, 0);                                                   \
    LOAD_B4(bf, BUF, 2);                                                      \
    ISSUE_HALF((T) + 2, 1, BUF);                                              \
    MFMA16(0);                                                                \
    /* phase 3 + tile-boundary wait.  The vmcnt must sit BEFORE this tile's \
       closing barrier: vmcnt is per-wave, and the next tile's ds_reads are \
       only safe once EVERY wave has drained its own staging DMA — wait,    \
       then rendezvous.  (A wait after the barrier lets a fast wave read    \
       slots whose glds were issued by a still-computing slow wave: seen as \
       nondeterministic corruption at 4k before the fix.) */                  \
    LOAD_A4(af, BUF, 3, 4);                                                   \
    ISSUE_HALF((T) + 2, 2, BUF);                                              \
    if (VAR == 0 || VAR == 2) __builtin_amdgcn_s_barrier();                   \
    if (VAR != 2 && VAR != 3) __builtin_amdgcn_s_setprio(1);                  \
    _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                          \
    _Pragma("unroll") for (int ni = 0; ni < 4; ++ni)                          \
        acc[4 + mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(            \
            af[mi], bf[ni], acc[4 + mi][ni], 0, 0, 0);                        \
    if (VAR != 2 && VAR != 3) __builtin_amdgcn_s_setprio(0);                  \
    if ((T) >= ntiles - 2)                                                    \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                        \
    else                                                                      \
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");                        \
    __builtin_amdgcn_s_barrier();                                             \
  } while (0)

  for (int t = 0; t < ntiles; t += 2) {
    TILE4(t, 0);
    TILE4(t + 1, 1);
  }
#undef TILE4
#undef MFMA16
#undef LOAD_A4
#undef LOAD_B4
#undef ISSUE_HALF

  // ---- epilogue: C-fragment map col = lane&15, row = (lane>>4)*4 + e ----
  const long long c_row0 = (long long)tile_m * PBM + wm * 128 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * PBN + wn * 64 + l15;
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
// Second-generation staging: FULL-tile stage units with 128-byte rows.
//
// The k-half slots above stage 64-B row segments (half a cacheline per
// request).  This body stages each operand TILE (256 rows x 64 k = 32 KiB)
// as one unit of 4 glds/thread with full 128-B-contiguous rows, halving the
// request count at equal bytes.  Phase-local consumption still holds:
//   reads:  P0 {A rows, kk0 chunks | B kk0}, P1 {A kk0 mi4..7},
//           P2 {kk1...}, P3 {...} — B unit last read P2-top (kk1 B frags),
//           A unit last read P3-top.
//   issue:  A(t+1) at P0 (into buf^1, old data dead since t-1 P3);
//           B(t+2) at P3 (into buf, B(t) last read P2-top).
//   wait:   vmcnt(4) at each tile boundary leaves only B(t+2)'s 4 loads in
//           flight -> A(t+1), B(t+1) landed.
// LDS swizzle for 128-B rows: chunk c of row r lands at (c + 2*(r>>1)) & 7
// — conflict-free for all four ds_read_b128 lane groups (verified against
// the (addr/4)%64 banking as before); kk=1 read offsets are kk=0's ^ 64 B.
// MODE: 1 = one barrier/phase + static young-half prio, bunched issues
//          (A(t+1) x4 at P0, B(t+2) x4 at P3), vmcnt(4);
//       2 = two barriers/phase + per-cluster setprio, bunched, vmcnt(4);
//       3 = like 1 but glds SPREAD one-or-two per phase (B(t+1)q2+A(t+1)q1
//           at P0, A q2 at P1, B(t+2)q1 at P3, its q2 at the next P0) with
//           vmcnt(2) boundaries;
//       4 = one barrier/phase + per-cluster setprio, bunched, vmcnt(4).
// ---------------------------------------------------------------------------
template <int MODE>
__device__ __forceinline__ void
gemm8ph_full_body(const ushort* __restrict__ A, const ushort* __restrict__ B,
                  ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / PBM) * (N / PBN);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / PBN;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;
  const int wn = wid & 3;
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;

  // slots: 0 = B tile, 1 = A tile (32 KiB each = 256 rows x 8 chunks)
  __shared__ __attribute__((aligned(16))) ushort lds[2][2][256 * 64];

  const ushort* Atile = A + (long long)tile_m * PBM * K;
  const ushort* Btile = B + (long long)tile_n * PBN * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // fragment read offset: row*128 + ((c + 2*(row>>1))&7)*16, c = kk*4+kgrp;
  // row = base + l15 with base % 16 == 0, so the swizzle is per-lane const
  // and the kk=1 offset is the kk=0 offset XOR 64.
  const int swz0 = (kgrp + 2 * ((l15 >> 1) & 3)) & 7;
  const int frag0 = l15 * 128 + swz0 * 16;               // bytes, kk = 0
  const int a_off = wm * 128 * 128 + frag0;              // + mi*16*128
  const int b_off = wn * 64 * 128 + frag0;               // + ni*16*128

  // staging source coords (i = 0..3): linear slot s = i*512 + tid covers
  // row = s>>3, dest chunk sc = s&7; source chunk = (sc - 2*(row>>1)) & 7
  int st_row[4], st_cofs[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int s = i * 512 + tid;
    const int r = s >> 3;
    const int c = ((s & 7) - 2 * ((r >> 1) & 3)) & 7;
    st_row[i] = r;
    st_cofs[i] = c * 8;  // elements
  }
  const int wave_chunk = tid & ~63;

  const int ntiles = K / PBK;

#define ISSUE_PART(TGT, SLOT, BUF, I0, NI)                                   \
  do {                                                                       \
    if ((TGT) < ntiles) {                                                    \
      const int k0_ = (TGT) * PBK;                                           \
      const ushort* op_ = (SLOT) ? Atile : Btile;                            \
      ushort* dst_ = &lds[(BUF)][(SLOT)][0];                                 \
      _Pragma("unroll") for (int i = (I0); i < (I0) + (NI); ++i) {           \
        const ushort* g = op_ + (long long)st_row[i] * K + k0_ + st_cofs[i]; \
        __builtin_amdgcn_global_load_lds(                                    \
            (const __attribute__((address_space(1))) unsigned int*)g,        \
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \
                (long long)(i * 512 + wave_chunk) * 8),                      \
            16, 0, 0);                                                       \
      }                                                                      \
    }                                                                        \
  } while (0)
#define ISSUE_TILE(TGT, SLOT, BUF) ISSUE_PART(TGT, SLOT, BUF, 0, 4)

  // prologue: B(0), A(0), then prime the next-tile stream
  ISSUE_TILE(0, 0, 0);
  ISSUE_TILE(0, 1, 0);
  if (MODE == 3) {
    ISSUE_PART(1, 0, 1, 0, 2);  // B(1) rows 0..127
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  } else {
    ISSUE_TILE(1, 0, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  if ((MODE == 1 || MODE == 3) &&
      __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // static young-half priority

#define LOAD_A4F(DST, BUF, MI0, KX)                                          \
  _Pragma("unroll") for (int j = 0; j < 4; ++j) {                            \
    DST[j] = *reinterpret_cast<const short8*>(                               \
        reinterpret_cast<const char*>(&lds[(BUF)][1][0]) +                   \
        ((a_off + ((MI0) + j) * 2048) ^ ((KX) * 64)));                       \
  }
#define LOAD_B4F(DST, BUF, KX)                                               \
  _Pragma("unroll") for (int j = 0; j < 4; ++j) {                            \
    DST[j] = *reinterpret_cast<const short8*>(                               \
        reinterpret_cast<const char*>(&lds[(BUF)][0][0]) +                   \
        ((b_off + j * 2048) ^ ((KX) * 64)));                                 \
  }
#define MFMA16F(MI0)                                                         \
  if (MODE == 2) __builtin_amdgcn_s_barrier();                               \
  if (MODE == 2 || MODE == 4) __builtin_amdgcn_s_setprio(1);                 \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                          \
  _Pragma("unroll") for (int ni = 0; ni < 4; ++ni)                          \
      acc[(MI0) + mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(         \
          af[mi], bf[ni], acc[(MI0) + mi][ni], 0, 0, 0);                     \
  if (MODE == 2 || MODE == 4) __builtin_amdgcn_s_setprio(0);                 \
  __builtin_amdgcn_s_barrier();

#define TILE4F(T, BUF)                                                       \
  do {                                                                       \
    short8 af[4], bf[4];                                                     \
    /* phase 0: A kk0 mi0..3 + B kk0; issue next-tile stages */              \
    LOAD_A4F(af, BUF, 0, 0);                                                 \
    LOAD_B4F(bf, BUF, 0);                                                    \
    if (MODE == 3) {                                                         \
      ISSUE_PART((T) + 1, 0, (BUF) ^ 1, 2, 2); /* B(t+1) rows 128..255 */    \
      ISSUE_PART((T) + 1, 1, (BUF) ^ 1, 0, 2); /* A(t+1) rows 0..127 */      \
    } else {                                                                 \
      ISSUE_TILE((T) + 1, 1, (BUF) ^ 1);                                     \
    }                                                                        \
    MFMA16F(0);                                                              \
    /* phase 1: A kk0 mi4..7 */                                              \
    LOAD_A4F(af, BUF, 4, 0);                                                 \
    if (MODE == 3) ISSUE_PART((T) + 1, 1, (BUF) ^ 1, 2, 2);                  \
    MFMA16F(4);                                                              \
    /* phase 2: A kk1 mi0..3 + B kk1 */                                      \
    LOAD_A4F(af, BUF, 0, 1);                                                 \
    LOAD_B4F(bf, BUF, 1);                                                    \
    MFMA16F(0);                                                              \
    /* phase 3: A kk1 mi4..7; issue B(T+2); tile-boundary wait BEFORE the  \
       closing barrier (vmcnt is per-wave: wait, then rendezvous — a wait  \
       after the barrier lets a fast wave read slots whose DMA a slow wave \
       has not yet drained; nondeterministic corruption at 4k before fix) */ \
    LOAD_A4F(af, BUF, 4, 1);                                                 \
    if (MODE == 3)                                                           \
      ISSUE_PART((T) + 2, 0, BUF, 0, 2);  /* B(t+2) rows 0..127 */           \
    else                                                                     \
      ISSUE_TILE((T) + 2, 0, BUF);                                           \
    if (MODE == 2) __builtin_amdgcn_s_barrier();                             \
    if (MODE == 2 || MODE == 4) __builtin_amdgcn_s_setprio(1);               \
    _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                         \
    _Pragma("unroll") for (int ni = 0; ni < 4; ++ni)                         \
        acc[4 + mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(           \
            af[mi], bf[ni], acc[4 + mi][ni], 0, 0, 0);                       \
    if (MODE == 2 || MODE == 4) __builtin_amdgcn_s_setprio(0);               \
    if ((T) >= ntiles - 2)                                                   \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                       \
    else if (MODE == 3)                                                      \
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");                       \
    else                                                                     \
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");                       \
    __builtin_amdgcn_s_barrier();                                            \
  } while (0)

  for (int t = 0; t < ntiles; t += 2) {
    TILE4F(t, 0);
    TILE4F(t + 1, 1);
  }
#undef TILE4F
#undef MFMA16F
#undef LOAD_A4F
#undef LOAD_B4F
#undef ISSUE_TILE
#undef ISSUE_PART

  const long long c_row0 = (long long)tile_m * PBM + wm * 128 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * PBN + wn * 64 + l15;
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

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                        ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_body<0>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v4_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_full_body<1>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v5_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_full_body<2>(A, B, C, M, N, K);
}

// ---------------------------------------------------------------------------
// Low-barrier generation: 2 phases per K-tile (or 1), full-tile staging,
// A double-buffered + B ring-of-3 across the WHOLE 160 KiB LDS.
//
// PMC on the 4-phase bodies shows 32% SQ_WAIT_ANY (parked at 4-8 barrier
// generations/tile) vs hipBLASLt's 8.6%.  This body cuts to 2 barriers/tile:
//   P0: {A(t) kk0 8x b128 + B(t) kk0 4x | issue A(t+1)->slot (t+1)&1 |
//        32 MFMA kk0 | BAR}
//   P1: {kk1 reads | issue B(t+2)->slot 2+(t+2)%3 | 32 MFMA kk1 |
//        vmcnt(4) | BAR}
// The B ring gives B(t+2) a 4-phase flight window; A(t+1) flies 1 full
// 32-MFMA phase (~1.1k cycles > HBM latency).  vmcnt(4) at the tile end
// leaves exactly B(t+2) in flight; everything the next tile reads is landed
// before the rendezvous (wait-then-barrier, as in the 4-phase fix).
// PHASES=1 merges both phases: one barrier/K-tile, counted vmcnt — the
// old structure minus its vmcnt(0) drain.
// ---------------------------------------------------------------------------
template <int PHASES, int GROUPED = 0, int PRIO = 1>
__device__ __forceinline__ void
gemm2ph_body(const ushort* __restrict__ A, const ushort* __restrict__ B,
             ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / PBM) * (N / PBN);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / PBN;
  int tile_m, tile_n;
  if (GROUPED) {
    // grouped-m flat order: each XCD's contiguous wgid chunk becomes a
    // GM-row x n 2D block, so its L2 holds GM A-panels + chunk/GM B-panels
    // instead of 1-2 A-panels + a whole row of B-panels.
    const int GM = 8;
    const int tiles_m = M / PBM;
    const int group = wgid / (GM * tiles_n);
    const int rem = wgid % (GM * tiles_n);
    const int g0 = group * GM;
    const int gh = (tiles_m - g0 < GM) ? (tiles_m - g0) : GM;
    tile_m = g0 + rem % gh;
    tile_n = rem / gh;
  } else {
    tile_m = wgid / tiles_n;
    tile_n = wgid % tiles_n;
  }

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;
  const int wn = wid & 3;
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;

  // 5 x 32 KiB slots = the full 160 KiB: A in 0,1 (dbuf); B ring in 2,3,4
  __shared__ __attribute__((aligned(16))) ushort lds[5][256 * 64];

  const ushort* Atile = A + (long long)tile_m * PBM * K;
  const ushort* Btile = B + (long long)tile_n * PBN * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int swz0 = (kgrp + 2 * ((l15 >> 1) & 3)) & 7;
  const int frag0 = l15 * 128 + swz0 * 16;
  const int a_off = wm * 128 * 128 + frag0;
  const int b_off = wn * 64 * 128 + frag0;

  int st_row[4], st_cofs[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int s = i * 512 + tid;
    const int r = s >> 3;
    const int c = ((s & 7) - 2 * ((r >> 1) & 3)) & 7;
    st_row[i] = r;
    st_cofs[i] = c * 8;
  }
  const int wave_chunk = tid & ~63;

  const int ntiles = K / PBK;

#define ISSUE2(TGT, OP, SLOT)                                                \
  do {                                                                       \
    if ((TGT) < ntiles) {                                                    \
      const int k0_ = (TGT) * PBK;                                           \
      ushort* dst_ = &lds[(SLOT)][0];                                        \
      _Pragma("unroll") for (int i = 0; i < 4; ++i) {                        \
        const ushort* g = (OP) + (long long)st_row[i] * K + k0_ + st_cofs[i];\
        __builtin_amdgcn_global_load_lds(                                    \
            (const __attribute__((address_space(1))) unsigned int*)g,        \
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \
                (long long)(i * 512 + wave_chunk) * 8),                      \
            16, 0, 0);                                                       \
      }                                                                      \
    }                                                                        \
  } while (0)

  // prologue: B(0)->2, A(0)->0, B(1)->3; wait all but B(1)
  ISSUE2(0, Btile, 2);
  ISSUE2(0, Atile, 0);
  ISSUE2(1, Btile, 3);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  if (PRIO && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // static young-half priority

#define LOAD_A8(DST, ASLOT, KX)                                              \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                            \
    DST[j] = *reinterpret_cast<const short8*>(                               \
        reinterpret_cast<const char*>(&lds[0][0]) + (ASLOT) * 32768 +        \
        ((a_off + j * 2048) ^ ((KX) * 64)));                                 \
  }
#define LOAD_B4R(DST, BSLOT, KX)                                             \
  _Pragma("unroll") for (int j = 0; j < 4; ++j) {                            \
    DST[j] = *reinterpret_cast<const short8*>(                               \
        reinterpret_cast<const char*>(&lds[0][0]) + (BSLOT) * 32768 +        \
        ((b_off + j * 2048) ^ ((KX) * 64)));                                 \
  }
#define MFMA32(KX)                                                           \
  _Pragma("unroll") for (int mi = 0; mi < 8; ++mi)                           \
  _Pragma("unroll") for (int ni = 0; ni < 4; ++ni)                           \
      acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                 \
          af[mi], bf[ni], acc[mi][ni], 0, 0, 0);

  for (int t = 0; t < ntiles; ++t) {
    const int aslot = t & 1;
    const int bslot = 2 + t % 3;
    const int bslot2 = 2 + (t + 2) % 3;
    short8 af[8], bf[4];
    // phase 0: kk0
    LOAD_A8(af, aslot, 0);
    LOAD_B4R(bf, bslot, 0);
    ISSUE2(t + 1, Atile, aslot ^ 1);
    MFMA32(0);
    if (PHASES == 2) __builtin_amdgcn_s_barrier();
    // phase 1: kk1
    LOAD_A8(af, aslot, 1);
    LOAD_B4R(bf, bslot, 1);
    ISSUE2(t + 2, Btile, bslot2);
    MFMA32(1);
    if (t >= ntiles - 2)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }
#undef MFMA32
#undef LOAD_A8
#undef LOAD_B4R
#undef ISSUE2

  const long long c_row0 = (long long)tile_m * PBM + wm * 128 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * PBN + wn * 64 + l15;
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

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v6_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_full_body<3>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v8_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm2ph_body<2>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v9_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm2ph_body<1>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v11_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                            ushort* __restrict__ C, int M, int N, int K) {
  gemm2ph_body<1, 1>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v12_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                            ushort* __restrict__ C, int M, int N, int K) {
  gemm2ph_body<1, 1, 0>(A, B, C, M, N, K);
}

// 16-wave (1024-thread) single-barrier body: 4 waves/SIMD, wave tile 64x64
// (4x4 fragments, 64 acc VGPRs) — rendezvous skew hides behind 4-way wave
// interleave on each SIMD.  Same 5-slot LDS (A dbuf + B ring-3), same
// swizzle; staging is 2 glds/thread per 32 KiB unit.
template <int PRIO>
__device__ __forceinline__ void
gemm2ph_wide_body(const ushort* __restrict__ A, const ushort* __restrict__ B,
                  ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / PBM) * (N / PBN);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / PBN;
  const int GM = 8;
  const int tiles_m = M / PBM;
  const int group = wgid / (GM * tiles_n);
  const int rem = wgid % (GM * tiles_n);
  const int g0 = group * GM;
  const int gh = (tiles_m - g0 < GM) ? (tiles_m - g0) : GM;
  const int tile_m = g0 + rem % gh;
  const int tile_n = rem / gh;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;           // 0..15
  const int wm = wid >> 2;            // 0..3 -> 64-row band of A
  const int wn = wid & 3;             // 0..3 -> 64-row band of B
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;

  __shared__ __attribute__((aligned(16))) ushort lds[5][256 * 64];

  const ushort* Atile = A + (long long)tile_m * PBM * K;
  const ushort* Btile = B + (long long)tile_n * PBN * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int swz0 = (kgrp + 2 * ((l15 >> 1) & 3)) & 7;
  const int frag0 = l15 * 128 + swz0 * 16;
  const int a_off = wm * 64 * 128 + frag0;
  const int b_off = wn * 64 * 128 + frag0;

  int st_row[2], st_cofs[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int s = i * 1024 + tid;
    const int r = s >> 3;
    const int c = ((s & 7) - 2 * ((r >> 1) & 3)) & 7;
    st_row[i] = r;
    st_cofs[i] = c * 8;
  }
  const int wave_chunk = tid & ~63;

  const int ntiles = K / PBK;

#define ISSUE2W(TGT, OP, SLOT)                                               \
  do {                                                                       \
    if ((TGT) < ntiles) {                                                    \
      const int k0_ = (TGT) * PBK;                                           \
      ushort* dst_ = &lds[(SLOT)][0];                                        \
      _Pragma("unroll") for (int i = 0; i < 2; ++i) {                        \
        const ushort* g = (OP) + (long long)st_row[i] * K + k0_ + st_cofs[i];\
        __builtin_amdgcn_global_load_lds(                                    \
            (const __attribute__((address_space(1))) unsigned int*)g,        \
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \
                (long long)(i * 1024 + wave_chunk) * 8),                     \
            16, 0, 0);                                                       \
      }                                                                      \
    }                                                                        \
  } while (0)

  ISSUE2W(0, Btile, 2);
  ISSUE2W(0, Atile, 0);
  ISSUE2W(1, Btile, 3);
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  if (PRIO && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 512)
    __builtin_amdgcn_s_setprio(1);  // static young-half priority

#define LOAD_W4(DST, SLOTBASE, OFF, KX)                                      \
  _Pragma("unroll") for (int j = 0; j < 4; ++j) {                            \
    DST[j] = *reinterpret_cast<const short8*>(                               \
        reinterpret_cast<const char*>(&lds[0][0]) + (SLOTBASE) * 32768 +     \
        (((OFF) + j * 2048) ^ ((KX) * 64)));                                 \
  }
#define MFMA16W(KX)                                                         \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                          \
  _Pragma("unroll") for (int ni = 0; ni < 4; ++ni)                          \
      acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                \
          af[mi], bf[ni], acc[mi][ni], 0, 0, 0);

  for (int t = 0; t < ntiles; ++t) {
    const int aslot = t & 1;
    const int bslot = 2 + t % 3;
    const int bslot2 = 2 + (t + 2) % 3;
    short8 af[4], bf[4];
    LOAD_W4(af, aslot, a_off, 0);
    LOAD_W4(bf, bslot, b_off, 0);
    ISSUE2W(t + 1, Atile, aslot ^ 1);
    MFMA16W(0);
    LOAD_W4(af, aslot, a_off, 1);
    LOAD_W4(bf, bslot, b_off, 1);
    ISSUE2W(t + 2, Btile, bslot2);
    MFMA16W(1);
    if (t >= ntiles - 2)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }
#undef MFMA16W
#undef LOAD_W4
#undef ISSUE2W

  const long long c_row0 = (long long)tile_m * PBM + wm * 64 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * PBN + wn * 64 + l15;
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

extern "C" __global__ void __launch_bounds__(1024, 4)
gemm_bt_bf16_8ph_v13_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                            ushort* __restrict__ C, int M, int N, int K) {
  gemm2ph_wide_body<1>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(1024, 4)
gemm_bt_bf16_8ph_v14_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                            ushort* __restrict__ C, int M, int N, int K) {
  gemm2ph_wide_body<0>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v7_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_full_body<4>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v1_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_body<1>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v2_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_body<2>(A, B, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_bf16_8ph_v3_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                           ushort* __restrict__ C, int M, int N, int K) {
  gemm8ph_body<3>(A, B, C, M, N, K);
}

// ---------------------------------------------------------------------------
// v16: TWO INDEPENDENT BLOCKS PER CU.  All prior bodies use >=128 KiB LDS
// (1 block/CU), so every rendezvous parks the whole CU (PMC: WAIT 26-36%).
// This body shrinks to BK=32 / 16 KiB stage units (A dbuf + B ring-3 =
// 80 KiB) and 512 threads at <=256 VGPR -> 2 blocks/CU of 8 waves each:
// the two blocks' barriers are independent, so one block's arrival skew
// hides behind the other block's MFMA stream.  One phase per K-tile:
// {A kk reads (8 b128) + B (4) | issue A(t+1), B(t+2) | 32 MFMA |
//  vmcnt(2) | BAR}.  Same 128-B-row staging + (c + 2*(row>>1))&7 swizzle
// (chunk c in 0..3 here: rows are 64 B... no — BK=32 bf16 = 64-B rows).
// NOTE rows are 64 B (4 chunks): the request-granularity cost that hurt
// v0-v3 at 8k returns; the bet is that 2-block overlap outweighs it.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(512, 2)
gemm_bt_bf16_8ph_v16_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                            ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / PBM) * (N / PBN);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / PBN;
  const int GM = 8;
  const int tiles_m = M / PBM;
  const int grp = wgid / (GM * tiles_n);
  const int rem = wgid % (GM * tiles_n);
  const int g0 = grp * GM;
  const int gh = (tiles_m - g0 < GM) ? (tiles_m - g0) : GM;
  const int tile_m = g0 + rem % gh;
  const int tile_n = rem / gh;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 2;            // 0..1 -> 128-row half of A
  const int wn = wid & 3;             // 0..3 -> 64-row band of B
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;         // chunk 0..3 (64-B rows)

  // 5 x 16 KiB slots = 80 KiB: A in 0,1; B ring in 2,3,4
  __shared__ __attribute__((aligned(16))) ushort lds[5][256 * 32];

  const ushort* Atile = A + (long long)tile_m * PBM * K;
  const ushort* Btile = B + (long long)tile_n * PBN * K;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // 64-B rows, 4 chunks: swizzle (c + 2*(row>>1)) & 3 spreads the two
  // same-parity rows of each ds_read_b128 lane-group class (<=2-way; the
  // 4-chunk space cannot reach the full 8-way spread of the 128-B layout)
  const int swz0 = (kgrp + 2 * ((l15 >> 1) & 1)) & 3;
  const int frag0 = l15 * 64 + swz0 * 16;
  const int a_off = wm * 128 * 64 + frag0;   // + mi*16*64
  const int b_off = wn * 64 * 64 + frag0;    // + ni*16*64

  unsigned st_off[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int s = i * 512 + tid;
    const int r = s >> 2;
    st_off[i] = (unsigned)r * (unsigned)K
                + (((((s & 3) - 2 * ((r >> 1) & 1)) & 3) * 8));
  }
  const int wave_chunk = tid & ~63;

  const int ntiles = K / 32;

#define ISSUE16(TGT, OP, SLOT)                                               \
  do {                                                                       \
    if ((TGT) < ntiles) {                                                    \
      const int k0_ = (TGT) * 32;                                            \
      const ushort* opk_ = (OP) + k0_;                                       \
      ushort* dst_ = &lds[(SLOT)][0];                                        \
      _Pragma("unroll") for (int i = 0; i < 2; ++i) {                        \
        const ushort* g = opk_ + st_off[i];                                  \
        __builtin_amdgcn_global_load_lds(                                    \
            (const __attribute__((address_space(1))) unsigned int*)g,        \
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \
                (long long)(i * 512 + wave_chunk) * 8),                      \
            16, 0, 0);                                                       \
      }                                                                      \
    }                                                                        \
  } while (0)

  ISSUE16(0, Btile, 2);
  ISSUE16(0, Atile, 0);
  ISSUE16(1, Btile, 3);
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  __builtin_amdgcn_s_barrier();

#define LOAD16(DST, SLOT, OFF, MI0, NFRAG)                                   \
  _Pragma("unroll") for (int j = 0; j < (NFRAG); ++j) {                      \
    DST[j] = *reinterpret_cast<const short8*>(                               \
        reinterpret_cast<const char*>(&lds[(SLOT)][0]) +                     \
        ((OFF) + ((MI0) + j) * 1024));                                       \
  }

  for (int t = 0; t < ntiles; ++t) {
    const int aslot = t & 1;
    const int bslot = 2 + t % 3;
    const int bslot2 = 2 + (t + 2) % 3;
    short8 af[8], bf[4];
    LOAD16(af, aslot, a_off, 0, 8);
    LOAD16(bf, bslot, b_off, 0, 4);
    ISSUE16(t + 1, Atile, aslot ^ 1);
    ISSUE16(t + 2, Btile, bslot2);
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    if (t >= ntiles - 2)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }
#undef LOAD16
#undef ISSUE16

  const long long c_row0 = (long long)tile_m * PBM + wm * 128 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * PBN + wn * 64 + l15;
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
// v17: the 2-blocks-per-CU premise in its proper geometry.  256x128 tile,
// 8 waves as 4M x 2N (wave tile 64x64 -> 64 acc VGPRs), BK=32 stage units
// (A 16 KiB dbuf + B 8 KiB ring-3 = 56 KiB LDS), __launch_bounds__(512, 4)
// capping at 128 VGPRs -> TWO independent blocks per CU: each block's
// per-tile vmcnt+barrier parks only ITSELF while the co-resident block's
// MFMA stream keeps the SIMDs fed (the single-block bodies park the whole
// CU: PMC WAIT 26-36%).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(512, 4)
gemm_bt_bf16_8ph_v17_kernel(const ushort* __restrict__ A, const ushort* __restrict__ B,
                            ushort* __restrict__ C, int M, int N, int K) {
  const int tiles_n = N / 128;
  const int tiles_m = M / 256;
  const int nwg = tiles_m * tiles_n;
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int GM = 8;
  const int grp = wgid / (GM * tiles_n);
  const int rem = wgid % (GM * tiles_n);
  const int g0 = grp * GM;
  const int gh = (tiles_m - g0 < GM) ? (tiles_m - g0) : GM;
  const int tile_m = g0 + rem % gh;
  const int tile_n = rem / gh;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1;            // 0..3 -> 64-row band of A
  const int wn = wid & 1;             // 0..1 -> 64-row band of B
  const int l15 = lane & 15;
  const int kgrp = lane >> 4;         // chunk 0..3 (64-B rows)

  // A slots 0,1: 256x32 (16 KiB); B ring slots at byte offsets after them:
  // one shared array (multiple __shared__ objects de-pipeline glds)
  __shared__ __attribute__((aligned(16))) ushort lds[(2 * 256 + 3 * 128) * 32];
#define V17_A(b) (&lds[(b) * 256 * 32])
#define V17_B(r) (&lds[2 * 256 * 32 + (r) * 128 * 32])

  const ushort* Atile = A + (long long)tile_m * 256 * K;
  const ushort* Btile = B + (long long)tile_n * 128 * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int swz0 = (kgrp + 2 * ((l15 >> 1) & 1)) & 3;
  const int frag0 = l15 * 64 + swz0 * 16;
  const int a_off = wm * 64 * 64 + frag0;   // + mi*16*64
  const int b_off = wn * 64 * 64 + frag0;   // + ni*16*64

  unsigned st_a[2];
  unsigned st_b;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int s = i * 512 + tid;
    const int r = s >> 2;
    st_a[i] = (unsigned)r * (unsigned)K
              + (((((s & 3) - 2 * ((r >> 1) & 1)) & 3) * 8));
  }
  {
    const int r = tid >> 2;
    st_b = (unsigned)r * (unsigned)K
           + (((((tid & 3) - 2 * ((r >> 1) & 1)) & 3) * 8));
  }
  const int wave_chunk = tid & ~63;

  const int ntiles = K / 32;

#define ISSUE17A(TGT, BUF)                                                   \
  do {                                                                       \
    if ((TGT) < ntiles) {                                                    \
      const ushort* opk_ = Atile + (TGT) * 32;                               \
      ushort* dst_ = V17_A(BUF);                                             \
      _Pragma("unroll") for (int i = 0; i < 2; ++i) {                        \
        __builtin_amdgcn_global_load_lds(                                    \
            (const __attribute__((address_space(1))) unsigned int*)(opk_ + st_a[i]), \
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \
                (long long)(i * 512 + wave_chunk) * 8),                      \
            16, 0, 0);                                                       \
      }                                                                      \
    }                                                                        \
  } while (0)
#define ISSUE17B(TGT, SLOT)                                                  \
  do {                                                                       \
    if ((TGT) < ntiles) {                                                    \
      const ushort* opk_ = Btile + (TGT) * 32;                               \
      ushort* dst_ = V17_B(SLOT);                                            \
      __builtin_amdgcn_global_load_lds(                                      \
          (const __attribute__((address_space(1))) unsigned int*)(opk_ + st_b), \
          (__attribute__((address_space(3))) unsigned int*)(dst_ +           \
              (long long)wave_chunk * 8),                                    \
          16, 0, 0);                                                         \
    }                                                                        \
  } while (0)

  ISSUE17B(0, 0);
  ISSUE17A(0, 0);
  ISSUE17B(1, 1);
  asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {
    const int aslot = t & 1;
    const int bslot = t % 3;
    const int bslot2 = (t + 2) % 3;
    short8 af[4], bf[4];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      af[j] = *reinterpret_cast<const short8*>(
          reinterpret_cast<const char*>(V17_A(aslot)) + a_off + j * 1024);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bf[j] = *reinterpret_cast<const short8*>(
          reinterpret_cast<const char*>(V17_B(bslot)) + b_off + j * 1024);
    ISSUE17A(t + 1, aslot ^ 1);
    ISSUE17B(t + 2, bslot2);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    if (t >= ntiles - 2)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }
#undef ISSUE17A
#undef ISSUE17B
#undef V17_A
#undef V17_B

  const long long c_row0 = (long long)tile_m * 256 + wm * 64 + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * 128 + wn * 64 + l15;
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
