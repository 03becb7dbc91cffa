// Paged decode attention (single new token per sequence) for gfx950.
//
// KV cache layout: [num_pages, PAGE_SIZE, Hk, D] bf16 — a token slot is
// contiguous over (Hk, D), so cache append is a single index_copy and a
// position's per-head K/V row is a contiguous 256 B read.
//
// v2 (round 2): the v1 kernel processed ONE position per wave-iteration
// with a full 6-step wave_reduce_sum in the serial chain, and its H x B
// grid filled 32 of 256 CUs at the bench's B=1 — a 27.6 us/layer latency
// floor at short contexts (profiles/r01_decode_breakdown2.txt).  v2 is
// flash-decoding proper:
//   - grid (H, B, NSPLIT=8): each block owns a context SLICE -> 256 blocks
//     fill the chip and the serial chain shrinks 8x;
//   - a wave processes 32 POSITIONS per pass: lane pairs (l, l+32) split
//     the D=128 dot (one shfl_xor combines), the softmax max/sum runs once
//     per 32-position batch instead of per position, and the V pass is one
//     broadcast + fma per position (no reduce in the chain);
//   - per-block (m, l, O) partials land in a scratch tensor; a second tiny
//     kernel merges the 8 slices per (b, h).  Both kernels are
//     graph-capturable (no host reads, stable shapes).

#include "common.h"

#define PAGE_SIZE 16
#define DA_NSPLIT 8

// partials layout: [B, H, NSPLIT, 2 + D] floats: m, l, O[128]
extern "C" __global__ void __launch_bounds__(256)
paged_decode_attn_partial_kernel(
    const ushort* __restrict__ Q,        // [B, H, D]
    const ushort* __restrict__ Kcache,   // [P, 16, Hk, D]
    const ushort* __restrict__ Vcache,   // [P, 16, Hk, D]
    float* __restrict__ partials,        // [B, H, NSPLIT, 2 + D]
    const int* __restrict__ block_table, // [B, max_pages]
    const int* __restrict__ ctx_lens,    // [B]
    int H, int Hk, int max_pages, float scale) {
  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const int split = blockIdx.z;
  const int D = 128;
  const int kvh = h / (H / Hk);
  const int ctx = ctx_lens[b];
  const int* pages = block_table + (long long)b * max_pages;
  float* part = partials + (((long long)b * H + h) * DA_NSPLIT + split) * (2 + D);

  const int chunk = (ctx + DA_NSPLIT - 1) / DA_NSPLIT;
  const int lo = split * chunk;
  const int hi = min(ctx, lo + chunk);

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int dhalf = lane >> 5;  // 0: d 0..63, 1: d 64..127

  // q half-row in registers (8 x bf16x8), scaled
  const ushort* qrow = Q + ((long long)b * H + h) * D + dhalf * 64;
  float qv[64];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(qrow + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) qv[i * 8 + j] = bf2f(v.v[j]) * scale;
  }

  float m = -INFINITY, l = 0.f, o0 = 0.f, o1 = 0.f;  // o: d = lane*2, +1

  // waves stride 32-position batches over [lo, hi)
  for (int p0 = lo + wid * 32; p0 < hi; p0 += 4 * 32) {
    const int p = p0 + l31;
    const bool ok = p < hi;
    const long long slot = ok
        ? (long long)pages[p / PAGE_SIZE] * PAGE_SIZE + (p % PAGE_SIZE) : 0;
    // dot(q, K[p]) over this lane's 64-elem half
    const ushort* krow = Kcache + (slot * Hk + kvh) * D + dhalf * 64;
    float s = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      bf16x8 kv = *reinterpret_cast<const bf16x8*>(krow + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) s += qv[i * 8 + j] * bf2f(kv.v[j]);
    }
    s += __shfl_xor(s, 32, 64);           // combine the two D-halves
    if (!ok) s = -INFINITY;
    // batch softmax over the 32 positions (lanes 0..31 hold them; 32..63 dup)
    float bmax = s;
#pragma unroll
    for (int d = 1; d < 32; d <<= 1) bmax = fmaxf(bmax, __shfl_xor(bmax, d, 64));
    const float m_new = fmaxf(m, bmax);
    const float alpha = (m == -INFINITY) ? 0.f : __expf(m - m_new);
    const float pw = (s == -INFINITY) ? 0.f : __expf(s - m_new);
    float bsum = pw;
#pragma unroll
    for (int d = 1; d < 32; d <<= 1) bsum += __shfl_xor(bsum, d, 64);
    l = l * alpha + bsum;
    o0 *= alpha;
    o1 *= alpha;
    m = m_new;
    // V pass: one broadcast + V-row fma per position (no reduce in chain)
#pragma unroll
    for (int j = 0; j < 32; ++j) {
      const float pj = __shfl(pw, j, 64);
      if (pj == 0.f) continue;
      const int pp = p0 + j;
      const long long vslot =
          (long long)pages[pp / PAGE_SIZE] * PAGE_SIZE + (pp % PAGE_SIZE);
      const ushort* vrow = Vcache + (vslot * Hk + kvh) * D;
      o0 += pj * bf2f(vrow[lane * 2]);
      o1 += pj * bf2f(vrow[lane * 2 + 1]);
    }
  }

  // merge the 4 waves' partials through LDS, wave 0 writes the slice partial
  __shared__ float sm[4], sl[4];
  __shared__ float so[4][256];
  sm[wid] = m;
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
    if (lane == 0) {
      part[0] = M;
      part[1] = L;
    }
    part[2 + lane * 2] = a0;
    part[2 + lane * 2 + 1] = a1;
  }
}

// merge NSPLIT slice partials per (b, h): one 64-thread wave per block
extern "C" __global__ void __launch_bounds__(64)
paged_decode_attn_merge_kernel(const float* __restrict__ partials,
                               ushort* __restrict__ O,  // [B, H, D]
                               int H) {
  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const int D = 128;
  const int lane = threadIdx.x;
  const float* base = partials + (((long long)b * H + h) * DA_NSPLIT) * (2 + D);

  float M = -INFINITY;
#pragma unroll
  for (int s = 0; s < DA_NSPLIT; ++s)
    M = fmaxf(M, base[s * (2 + D)]);
  float L = 0.f, a0 = 0.f, a1 = 0.f;
#pragma unroll
  for (int s = 0; s < DA_NSPLIT; ++s) {
    const float ms = base[s * (2 + D)];
    if (ms == -INFINITY) continue;
    const float f = __expf(ms - M);
    L += base[s * (2 + D) + 1] * f;
    a0 += base[s * (2 + D) + 2 + lane * 2] * f;
    a1 += base[s * (2 + D) + 2 + lane * 2 + 1] * f;
  }
  const float inv = (L > 0.f) ? 1.f / L : 0.f;
  ushort* orow = O + ((long long)b * H + h) * D;
  orow[lane * 2] = f2bf(a0 * inv);
  orow[lane * 2 + 1] = f2bf(a1 * inv);
}
