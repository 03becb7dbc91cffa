#include "hip/hip_runtime.h"
// Grouped GEMM for MoE (Mixtral-style) on gfx950.
//
// After top-k routing, tokens are sorted by expert into A_sorted and each
// expert e owns rows [seg_starts[e], seg_starts[e+1]).  One launch computes
// every expert's C_seg = A_seg @ W_e^T: the host builds a flat tile map of
// (expert, m_tile) pairs (so no idle blocks for empty experts — router
// imbalance costs nothing extra), and each 256-thread block runs the same
// 128x128 glds-double-buffered MFMA pipeline as the dense GEMM, with
// per-row store masking at segment boundaries.
//
// W: [E, N, K] bf16 row-major.  A_sorted: [T_pad, N? no: K].  C: [T_pad, N].

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define GBK 64

__device__ __forceinline__ void g_stage_tile(
    const ushort* __restrict__ src, long long ldK, ushort* lds_tile, int tid) {
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

__device__ __forceinline__ short8 g_read_frag(const ushort* lds_tile, int row, int c) {
  const int slot = row * 8 + (c ^ (row & 7));
  return *reinterpret_cast<const short8*>(lds_tile + slot * 8);
}

extern "C" __global__ void __launch_bounds__(256, 2)
grouped_gemm_bt_bf16_kernel(const ushort* __restrict__ A,  // [T_pad, K] sorted by expert
                            const ushort* __restrict__ W,  // [E, N, K]
                            ushort* __restrict__ C,        // [T_pad, N]
                            const int* __restrict__ tile_expert,  // [n_mtiles]
                            const int* __restrict__ tile_m0,      // [n_mtiles] row of tile start
                            const int* __restrict__ seg_ends,     // [E]
                            int N, int K, int tiles_n) {
  const int mt = blockIdx.x / tiles_n;
  const int tile_n = blockIdx.x % tiles_n;
  const int e = tile_expert[mt];
  const int m0 = tile_m0[mt];
  const int seg_end = seg_ends[e];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1;
  const int wn = wid & 1;

  __shared__ __attribute__((aligned(16))) ushort lds[2][2][128 * GBK];

  const ushort* Atile = A + (long long)m0 * K;
  const ushort* Btile = W + ((long long)e * N + (long long)tile_n * 128) * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int m_base = wm * 64;
  const int n_base = wn * 64;
  const int frag_row = lane & 15;
  const int frag_kgrp = lane >> 4;

  const int ntiles = K / GBK;
  g_stage_tile(Atile, K, lds[0][0], tid);
  g_stage_tile(Btile, K, lds[0][1], tid);
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      g_stage_tile(Atile + (long long)(t + 1) * GBK, K, lds[buf ^ 1][0], tid);
      g_stage_tile(Btile + (long long)(t + 1) * GBK, K, lds[buf ^ 1][1], tid);
    }
    const ushort* Al = lds[buf][0];
    const ushort* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      short8 af[4], bf[4];
      const int c = kk * 4 + frag_kgrp;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = g_read_frag(Al, m_base + mi * 16 + frag_row, c);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = g_read_frag(Bl, n_base + ni * 16 + frag_row, c);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    buf ^= 1;
  }

  // epilogue with segment-boundary row masking
  const long long c_col0 = (long long)tile_n * 128 + n_base + (lane & 15);
  const int row_in_tile0 = m_base + (lane >> 4) * 4;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int e4 = 0; e4 < 4; ++e4) {
      const long long row = (long long)m0 + row_in_tile0 + mi * 16 + e4;
      if (row >= seg_end) continue;
      ushort* crow = C + row * N;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        crow[c_col0 + ni * 16] = f2bf(acc[mi][ni][e4]);
    }
  }
}

// ---------------------------------------------------------------------------
// Routed-FFN combine: out[t] = sum_j w[pos(t,j)] * down[pos(t,j)]  (bf16).
// pos = inverse routing permutation [T, k] (each token's k rows in the
// expert-sorted layout), so every output row is a private k-way sum — no
// atomics, one pass, replacing zeros + f32 index_add_ + cast (three torch
// kernels, ~6% of the Mixtral step).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
moe_combine_kernel(const ushort* __restrict__ down, const int* __restrict__ pos,
                   const float* __restrict__ weight, ushort* __restrict__ out,
                   int H, int k) {
  const long long t = blockIdx.x;
  const int nvec = H / 8;
  // per-token row indices + weights are wave-uniform scalars
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.f;
    for (int j = 0; j < k; ++j) {
      const int p = pos[t * k + j];
      const float w = weight[p];
      bf16x8 v = *reinterpret_cast<const bf16x8*>(down + (long long)p * H + i * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += w * bf2f(v.v[e]);
    }
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) o.v[e] = f2bf(acc[e]);
    *reinterpret_cast<bf16x8*>(out + t * H + i * 8) = o;
  }
}
