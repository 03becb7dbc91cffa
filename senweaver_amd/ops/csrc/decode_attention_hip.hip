#include "hip/hip_runtime.h"
// Paged decode attention (single new token per sequence) for gfx950.
//
// KV cache layout: [num_pages, PAGE_SIZE, Hk, D] bf16 — a token slot is
// contiguous over (Hk, D), so cache append is a single index_copy and a
// position's per-head K/V row is a contiguous 256 B read.
//
// One 256-thread block per (b, h): 4 waves stride the context positions
// (flash-decoding style), each wave keeps online-softmax state (m, l) and a
// 128-wide f32 O partial (2 elements/lane); partials merge through LDS at
// the end.  Memory-bound by K/V reads (512 B/position) — the right regime
// for batch<=beam decode, no MFMA needed.

#include "common.h"

#define PAGE_SIZE 16

extern "C" __global__ void __launch_bounds__(256)
paged_decode_attn_kernel(const ushort* __restrict__ Q,        // [B, H, D]
                         const ushort* __restrict__ Kcache,   // [P, 16, Hk, D]
                         const ushort* __restrict__ Vcache,   // [P, 16, Hk, D]
                         ushort* __restrict__ O,              // [B, H, D]
                         const int* __restrict__ block_table, // [B, max_pages]
                         const int* __restrict__ ctx_lens,    // [B]
                         int H, int Hk, int max_pages, float scale) {
  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const int D = 128;
  const int kvh = h / (H / Hk);
  const int ctx = ctx_lens[b];
  const int* pages = block_table + (long long)b * max_pages;

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;

  // q in registers: 2 elements per lane (scaled)
  const ushort* qrow = Q + ((long long)b * H + h) * D;
  const float q0 = bf2f(qrow[lane * 2]) * scale;
  const float q1 = bf2f(qrow[lane * 2 + 1]) * scale;

  float m = -INFINITY, l = 0.f, o0 = 0.f, o1 = 0.f;

  for (int p = wid; p < ctx; p += 4) {
    const long long slot = (long long)pages[p / PAGE_SIZE] * PAGE_SIZE + (p % PAGE_SIZE);
    const ushort* krow = Kcache + (slot * Hk + kvh) * D;
    const ushort* vrow = Vcache + (slot * Hk + kvh) * D;
    // dot(q, k): 2 elements per lane, wave reduce
    float s = q0 * bf2f(krow[lane * 2]) + q1 * bf2f(krow[lane * 2 + 1]);
    s = wave_reduce_sum(s);
    const float m_new = fmaxf(m, s);
    const float alpha = (m == -INFINITY) ? 0.f : __expf(m - m_new);
    const float pw = __expf(s - m_new);
    l = l * alpha + pw;
    o0 = o0 * alpha + pw * bf2f(vrow[lane * 2]);
    o1 = o1 * alpha + pw * bf2f(vrow[lane * 2 + 1]);
    m = m_new;
  }

  // merge the 4 waves' (m, l, o) partials
  __shared__ float sm[4], sl[4];
  __shared__ float so[4][256];  // [wave][2 per lane]
  sm[wid] = m;  // wave-uniform
  sl[wid] = l;
  so[wid][lane * 2] = o0;
  so[wid][lane * 2 + 1] = o1;
  __syncthreads();
  if (wid == 0) {
    float M = -INFINITY;
#pragma unroll
    for (int w = 0; w < 4; ++w) M = fmaxf(M, sm[w]);
    float L = 0.f, a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      if (sm[w] == -INFINITY) continue;
      const float f = __expf(sm[w] - M);
      L += sl[w] * f;
      a0 += so[w][lane * 2] * f;
      a1 += so[w][lane * 2 + 1] * f;
    }
    const float inv = (L > 0.f) ? 1.f / L : 0.f;
    ushort* orow = O + ((long long)b * H + h) * D;
    orow[lane * 2] = f2bf(a0 * inv);
    orow[lane * 2 + 1] = f2bf(a1 * inv);
  }
}
