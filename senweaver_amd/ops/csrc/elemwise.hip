// Fused elementwise / normalization kernels for the Llama backbone (gfx950).
//
// All memory-bound: target the HBM roofline (~6.3 TB/s achievable on MI355X),
// so every bf16 access is vectorized as 16 B/lane (bf16x8) per the CDNA4
// guideline (scalar bf16 loads cost ~2x).  The reference framework has no GPU
// math at all (its LLM is a remote HTTPS provider); these
// kernels are the MI355X-native compute path that replaces it.

#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm (optionally fused with residual add).
//   residual != nullptr:  r = x + residual  (written back to residual buffer,
//                         becoming the next layer's residual stream)
//                         y = r * rsqrt(mean(r^2) + eps) * w
// One workgroup (256 threads) per row; f32 accumulation.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
rmsnorm_fwd_kernel(const ushort* __restrict__ x, const ushort* __restrict__ w,
                   ushort* __restrict__ y, ushort* __restrict__ residual,
                   int hidden, float eps) {
  const int row = blockIdx.x;
  const long long base = (long long)row * hidden;
  const int nvec = hidden / 8;  // hidden % 8 == 0 enforced host-side
  const bool fuse_res = residual != nullptr;

  // Pass 1: sum of squares (with fused residual add written back; pass 2
  // re-reads from L1/L2 — a register cache here would be runtime-indexed and
  // spill to scratch, which is slower than the cache hit)
  float ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + base + i * 8);
    if (fuse_res) {
      bf16x8 r = *reinterpret_cast<const bf16x8*>(residual + base + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = f2bf(bf2f(v.v[j]) + bf2f(r.v[j]));
      *reinterpret_cast<bf16x8*>(residual + base + i * 8) = v;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v.v[j]);
      ss += f * f;
    }
  }

  // Block reduction
  __shared__ float smem[8];
  float wsum = wave_reduce_sum(ss);
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) smem[wid] = wsum;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < 4; ++i) total += smem[i];
  const float inv_rms = rsqrtf(total / (float)hidden + eps);

  const ushort* src = fuse_res ? residual : x;
  // Pass 2: scale
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(src + base + i * 8);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i * 8);
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      out.v[j] = f2bf(bf2f(v.v[j]) * inv_rms * bf2f(wv.v[j]));
    *reinterpret_cast<bf16x8*>(y + base + i * 8) = out;
  }
}

// ---------------------------------------------------------------------------
// RoPE (Llama / NeoX rotate-half style), fused over Q and K.
//   q: [T, Hq, D], k: [T, Hk, D] bf16, modified in place.
//   cos_sin: [max_pos, D] f32 laid out [cos(0..D/2) | sin(0..D/2)] per row
//   positions: [T] int32
// Precomputed host-side trig table (on-device sinf/cosf turns this
// memory-bound op VALU-bound).  Grid: (T, Hq+Hk).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(64)
rope_fwd_kernel(ushort* __restrict__ q, ushort* __restrict__ k,
                const float* __restrict__ cos_sin,
                const int* __restrict__ positions, int Hq, int Hk, int D) {
  const int t = blockIdx.x;
  const int h = blockIdx.y;
  const int half = D / 2;
  ushort* ptr;
  if (h < Hq) {
    ptr = q + ((long long)t * Hq + h) * D;
  } else {
    ptr = k + ((long long)t * Hk + (h - Hq)) * D;
  }
  const float* cs = cos_sin + (long long)positions[t] * D;
  // D = 128: 64 lanes each handle one rotation pair (d, d+half)
  for (int d = threadIdx.x; d < half; d += blockDim.x) {
    float c = cs[d];
    float s = cs[half + d];
    float x1 = bf2f(ptr[d]);
    float x2 = bf2f(ptr[d + half]);
    ptr[d] = f2bf(x1 * c - x2 * s);
    ptr[d + half] = f2bf(x2 * c + x1 * s);
  }
}

// ---------------------------------------------------------------------------
// Fused RoPE + head-major scatter: reads q/k in token-major [T, H, D] (the
// qkv GEMM's natural layout), applies rotate-half RoPE, and writes the
// attention kernel's [B, H, S, D] layout directly — replacing a rope pass
// plus two transpose copies (one full q/k read+write each) with one pass.
// Grid: (T, ceil((Hq+Hk)/heads_per_blk)), 256 threads covering several heads.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
rope_scatter_kernel(const ushort* __restrict__ qkv, long long ld,
                    ushort* __restrict__ q_out, ushort* __restrict__ k_out,
                    const float* __restrict__ cos_sin,
                    const int* __restrict__ positions, int Hq, int Hk, int D,
                    int S) {
  // reads q/k heads straight out of the fused qkv GEMM output (row stride
  // ld = q_size + 2*kv_size) — no separate contiguous slice copies.
  // 256-thread blocks cover (256 / (D/2)) heads each: adjacent heads read
  // adjacent qkv segments, so loads coalesce across the whole block (the
  // one-head-per-64-thread-block version ran ~1.9 TB/s in situ).
  const int t = blockIdx.x;
  const int half = D / 2;
  const int lanes_per_head = half / 8;     // bf16x8-vectorized: 8 d per lane
  const int heads_per_blk = 256 / lanes_per_head;
  const int h = blockIdx.y * heads_per_blk + threadIdx.x / lanes_per_head;
  if (h >= Hq + Hk) return;
  const int b = t / S;
  const int s = t % S;
  const ushort* src;
  ushort* dst;
  if (h < Hq) {
    src = qkv + (long long)t * ld + (long long)h * D;
    dst = q_out + (((long long)b * Hq + h) * S + s) * D;
  } else {
    src = qkv + (long long)t * ld + (long long)Hq * D + (long long)(h - Hq) * D;
    dst = k_out + (((long long)b * Hk + (h - Hq)) * S + s) * D;
  }
  const float* cs = cos_sin + (long long)positions[t] * D;
  const int d0 = (threadIdx.x % lanes_per_head) * 8;
  // 16-B vector loads/stores (scalar bf16 stores ran this at ~2 TB/s)
  bf16x8 x1 = *reinterpret_cast<const bf16x8*>(src + d0);
  bf16x8 x2 = *reinterpret_cast<const bf16x8*>(src + d0 + half);
  bf16x8 o1, o2;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float c = cs[d0 + j];
    const float sn = cs[half + d0 + j];
    const float a = bf2f(x1.v[j]);
    const float bb = bf2f(x2.v[j]);
    o1.v[j] = f2bf(a * c - bb * sn);
    o2.v[j] = f2bf(bb * c + a * sn);
  }
  *reinterpret_cast<bf16x8*>(dst + d0) = o1;
  *reinterpret_cast<bf16x8*>(dst + d0 + half) = o2;
}

// ---------------------------------------------------------------------------
// SwiGLU activation: y = silu(gate) * up, bf16, vectorized.
//   gateup: [T, 2*I] (fused gate|up GEMM output), y: [T, I]
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
swiglu_fwd_kernel(const ushort* __restrict__ gateup, ushort* __restrict__ y,
                  long long rows, int inter) {
  // 2D grid (vec-chunks, rows): no div/mod per element — the grid-stride
  // version's 64-bit divide was throttling this to ~1.4 TB/s in situ
  const int nvec = inter / 8;
  const long long r = blockIdx.y;
  const int i = blockIdx.x * 256 + threadIdx.x;
  if (i >= nvec) return;
  bf16x8 g = *reinterpret_cast<const bf16x8*>(gateup + r * 2 * inter + i * 8);
  bf16x8 u = *reinterpret_cast<const bf16x8*>(gateup + r * 2 * inter + inter + i * 8);
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float gf = bf2f(g.v[j]);
    float uf = bf2f(u.v[j]);
    float silu = gf / (1.f + __expf(-gf));
    out.v[j] = f2bf(silu * uf);
  }
  *reinterpret_cast<bf16x8*>(y + r * inter + i * 8) = out;
}

// ---------------------------------------------------------------------------
// Residual add (bf16): y = a + b  (used where the fused rmsnorm path isn't)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
add_bf16_kernel(const ushort* __restrict__ a, const ushort* __restrict__ b,
                ushort* __restrict__ y, long long n8) {
  for (long long idx = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       idx < n8; idx += (long long)gridDim.x * blockDim.x) {
    bf16x8 va = *reinterpret_cast<const bf16x8*>(a + idx * 8);
    bf16x8 vb = *reinterpret_cast<const bf16x8*>(b + idx * 8);
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out.v[j] = f2bf(bf2f(va.v[j]) + bf2f(vb.v[j]));
    *reinterpret_cast<bf16x8*>(y + idx * 8) = out;
  }
}

// ---------------------------------------------------------------------------
// V^T extraction: qkv [T, ld] (v slice at v_off) -> vt [B, Hk, D, S].
// LDS-tiled 32x32 transpose: reads are contiguous along d, writes contiguous
// along s — replaces a strided torch permute().contiguous() that ran ~4x off
// bandwidth.  Guards handle S not a multiple of 32.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
transpose_v_kernel(const ushort* __restrict__ qkv, long long ld, long long v_off,
                   ushort* __restrict__ vt, int B, int S, int Hk, int D) {
  // 16-B vector loads AND stores (the elementwise version's 2-B scalar
  // stores ran ~0.9 TB/s).  32 s-rows x 64 d-cols per block keeps all 256
  // threads busy in both phases; row pad of 8 (72-ushort stride) preserves
  // vector alignment and spreads rows over bank groups.
  __shared__ ushort tile[32][72];
  const int bh = blockIdx.z;       // b * Hk + hk
  const int b = bh / Hk;
  const int hk = bh % Hk;
  const int s0 = blockIdx.x * 32;
  const int d0 = blockIdx.y * 64;
  const ushort* src = qkv + (long long)b * S * ld + v_off + (long long)hk * D;
  {
    const int row = threadIdx.x >> 3;       // 0..31 along s
    const int sub = threadIdx.x & 7;        // 8 chunks of 8 along d
    const int s = s0 + row;
    const int d = d0 + sub * 8;
    if (s < S && d < D) {
      if (d + 7 < D) {
        *reinterpret_cast<bf16x8*>(&tile[row][sub * 8]) =
            *reinterpret_cast<const bf16x8*>(src + (long long)s * ld + d);
      } else {
        for (int j = 0; j < 8 && d + j < D; ++j)
          tile[row][sub * 8 + j] = src[(long long)s * ld + d + j];
      }
    }
  }
  __syncthreads();
  ushort* dst = vt + (((long long)b * Hk + hk) * D) * S;
  {
    const int row = threadIdx.x >> 2;       // 0..63 along d
    const int sub = threadIdx.x & 3;        // 4 chunks of 8 along s
    const int d = d0 + row;
    const int s = s0 + sub * 8;
    if (d < D && s < S) {
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o.v[j] = tile[sub * 8 + j][row];
      if (s + 7 < S) {
        *reinterpret_cast<bf16x8*>(dst + (long long)d * S + s) = o;
      } else {
        for (int j = 0; j < 8 && s + j < S; ++j)
          dst[(long long)d * S + s + j] = o.v[j];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Decode-step fused head dispatch: reads the fused qkv projection row
// directly (strided slices), applies rotate-half RoPE to q/k, and writes
//   q  -> q_out [B, Hq, D] (contiguous, ready for paged attention)
//   k  -> kcache[slot[b]] rows, v -> vcache[slot[b]] rows
// One kernel replacing SIX per decode layer (3 slice-contiguous copies +
// rope_inplace + 2 cache index_copy), each of which sat at the ~5 us
// small-kernel floor (profiles/r01_decode_breakdown.txt).
// Caches viewed as [P*16, Hk, D]; 16-B/lane vectorized.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
rope_kv_append_kernel(const ushort* __restrict__ qkv, long long ld,
                      ushort* __restrict__ q_out, ushort* __restrict__ kcache,
                      ushort* __restrict__ vcache,
                      const float* __restrict__ cos_sin,
                      const int* __restrict__ positions,
                      const int* __restrict__ slot,
                      int Hq, int Hk, int D) {
  const int b = blockIdx.x;
  const int half = D / 2;
  const int lanes_per_head = half / 8;
  const int heads_per_blk = 256 / lanes_per_head;
  const int h = blockIdx.y * heads_per_blk + threadIdx.x / lanes_per_head;
  const int H_total = Hq + 2 * Hk;
  if (h >= H_total) return;
  const int d0 = (threadIdx.x % lanes_per_head) * 8;

  const ushort* src;
  ushort* dst;
  bool do_rope = true;
  if (h < Hq) {
    src = qkv + (long long)b * ld + (long long)h * D;
    dst = q_out + ((long long)b * Hq + h) * D;
  } else if (h < Hq + Hk) {
    src = qkv + (long long)b * ld + (long long)(Hq + (h - Hq)) * D;
    dst = kcache + ((long long)slot[b] * Hk + (h - Hq)) * D;
  } else {
    src = qkv + (long long)b * ld + (long long)(Hq + Hk + (h - Hq - Hk)) * D;
    dst = vcache + ((long long)slot[b] * Hk + (h - Hq - Hk)) * D;
    do_rope = false;
  }
  if (!do_rope) {
    // v: plain 16-B copies of both halves' chunks
    *reinterpret_cast<bf16x8*>(dst + d0) =
        *reinterpret_cast<const bf16x8*>(src + d0);
    *reinterpret_cast<bf16x8*>(dst + d0 + half) =
        *reinterpret_cast<const bf16x8*>(src + d0 + half);
    return;
  }
  const float* cs = cos_sin + (long long)positions[b] * D;
  bf16x8 x1 = *reinterpret_cast<const bf16x8*>(src + d0);
  bf16x8 x2 = *reinterpret_cast<const bf16x8*>(src + d0 + half);
  bf16x8 o1, o2;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float c = cs[d0 + j];
    const float sn = cs[half + d0 + j];
    const float a = bf2f(x1.v[j]);
    const float bb = bf2f(x2.v[j]);
    o1.v[j] = f2bf(a * c - bb * sn);
    o2.v[j] = f2bf(bb * c + a * sn);
  }
  *reinterpret_cast<bf16x8*>(dst + d0) = o1;
  *reinterpret_cast<bf16x8*>(dst + d0 + half) = o2;
}
