// Flash attention forward v2 (gfx950): swapped 32x32 QK^T + in-register
// softmax — the CDNA4-guide structure that removes v1's dominant cost
// (PMC: 55% of wave-cycles parked on LDS round-trips and 4-level shuffle
// reduction chains in the softmax).
//
// Key ideas:
//   - compute S^T = mfma_32x32x16(K_tile, Q): for x^T y both operands are
//     plain 16-byte ROW fragments (A: K rows from LDS; B: Q rows from
//     registers) — the reduction axis (kv) becomes register-local per q
//     column, so the row max/sum is 31 in-lane ops + ONE cross-half shuffle;
//   - P^T stays in registers: cvt_pk_bf16 pairs + v_permlane32_swap build
//     the PV B-fragments directly (T12), no LDS P tile, no barriers between
//     QK and PV;
//   - PV accumulates O^T[d][q] (A = V^T rows from LDS); every O register
//     belongs to the lane's single q column, so the online-softmax rescale
//     is a plain 64-register scale;
//   - epilogue stores O^T coalesced into a [B, Hq, D, S] tensor (lanes =
//     consecutive q positions).
//
// Geometry: 256-thread block = 4 waves x 32 q rows = 128 q rows/block;
// KV tiles of 64 (two 32-kv subtiles); K LDS [64][128] + VT LDS [128][64]
// (32 KiB, no double buffer — occupancy does the hiding: ~150 VGPR ->
// 3 waves/SIMD).  D = 128, causal, GQA.  S % 128 == 0.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define V2_KVBLK 64
#define V2_QBLK 128  // per block; 32 per wave

// stage a [rows x row_bf16] tile into lane-linear LDS with the (chunk ^ row)
// XOR source swizzle (16-byte chunks; chunks_per_row 16 for K, 8 for VT)
__device__ __forceinline__ void v2_stage(const ushort* __restrict__ src,
                                         long long ld, int chunks_per_row,
                                         ushort* lds_tile, int chunks_total,
                                         int tid) {
  // XOR swizzle over the FULL chunk index (mask = chunks_per_row-1): a
  // 16-lane ds_read_b128 phase reads 16 consecutive rows at one logical
  // chunk, and a full-width XOR maps them to 16 distinct bank groups
  // (the 3-bit XOR left rows r and r+8 on the same banks: 2-way conflict,
  // PMC SQ_LDS_BANK_CONFLICT ~1.1e9 cycles at the bench shape)
  const int cmask = chunks_per_row - 1;
  const int wave_chunk = tid & ~63;
  for (int s0 = 0; s0 < chunks_total; s0 += 256) {
    const int s = s0 + tid;
    const int row = s / chunks_per_row;
    const int cl = s % chunks_per_row;
    const int c = (cl ^ row) & cmask;
    const ushort* g = src + (long long)row * ld + c * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_tile + ((long long)(s0 + wave_chunk)) * 8),
        16, 0, 0);
  }
}

__device__ __forceinline__ short8 v2_read(const ushort* lds_tile, int row,
                                          int c, int chunks_per_row) {
  const int phys = (c ^ row) & (chunks_per_row - 1);
  return *reinterpret_cast<const short8*>(lds_tile + (row * chunks_per_row + phys) * 8);
}

// v_cvt_pk_bf16_f32 has no builtin on gfx950 (guide T12)
__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

extern "C" __global__ void __launch_bounds__(256)
attn_fwd_v2_kernel(const ushort* __restrict__ Q, const ushort* __restrict__ K,
                   const ushort* __restrict__ VT, ushort* __restrict__ OT,
                   int B, int H, int Hk, int S, float scale) {
  // heaviest-first: causal work grows with the q-block index
  const int qb = gridDim.x - 1 - blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (H / Hk);
  const int D = 128;

  const ushort* Qh = Q + (((long long)b * H + h) * S) * D;
  const ushort* Kh = K + (((long long)b * Hk + kvh) * S) * D;
  const ushort* VTh = VT + (((long long)b * Hk + kvh) * D) * S;
  ushort* OTh = OT + (((long long)b * H + h) * D) * S;  // [D][S]

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int lhi = lane >> 5;           // 0/1 half
  const int q0 = qb * V2_QBLK + wid * 32;
  const int q_lane = q0 + l31;         // this lane's q row (shared with lane^32)

  __shared__ __attribute__((aligned(16))) ushort smem[V2_KVBLK * 128 + 128 * V2_KVBLK];
  ushort* k_lds = smem;
  ushort* vt_lds = smem + V2_KVBLK * 128;

  // Q fragments: lane holds Q[q_lane][kg*8..+7] for 16 chunks (kg pattern
  // (l>>5)*8+e per mfma step; we keep all 8 16-B chunks of the row half
  // this lane needs: chunk index = step*2 + lhi over 8 K-steps of 16)
  short8 qf[8];
  {
#pragma unroll
    for (int st = 0; st < 8; ++st)
      qf[st] = *reinterpret_cast<const short8*>(Qh + (long long)q_lane * D + st * 16 + lhi * 8);
  }

  float m_run = -INFINITY, l_run = 0.f;
  f32x16 o_acc[4];  // O^T: 4 d-blocks of 32 rows, col = q_lane
#pragma unroll
  for (int db = 0; db < 4; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[db][r] = 0.f;

  const int kv_end = min(S, qb * V2_QBLK + V2_QBLK);
  for (int kv0 = 0; kv0 < kv_end; kv0 += V2_KVBLK) {
    v2_stage(Kh + (long long)kv0 * D, D, 16, k_lds, V2_KVBLK * 16, tid);
    v2_stage(VTh + kv0, S, 8, vt_lds, 128 * 8, tid);
    __syncthreads();

    // ---- S^T[kv64][q32] via two 32x32 subtiles ----
    __builtin_amdgcn_s_setprio(1);
    f32x16 st[2];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
      for (int stp = 0; stp < 8; ++stp) {  // D=128 in 8 K=16 steps
        // A = K rows: lane reads K[sub*32 + l31][stp*16 + lhi*8 ..+7]
        short8 kf = v2_read(k_lds, sub * 32 + l31, stp * 2 + lhi, 16);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], acc, 0, 0, 0);
      }
      st[sub] = acc;
    }

    // ---- causal mask + scale + in-register online softmax ----
    // lane's values: S^T[kv = sub*32 + (r&3)+8*(r>>2)+4*lhi][q_lane]
    float vals[32];
    float tile_max = -INFINITY;
#pragma unroll
    for (int sub = 0; sub < 2; ++sub)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
        float v = st[sub][r] * scale;
        if (kv > q_lane) v = -INFINITY;
        vals[sub * 16 + r] = v;
        tile_max = fmaxf(tile_max, v);
      }
    __builtin_amdgcn_s_setprio(0);
    // combine the lane pair (other 32 kv of this q) — ONE cross-lane op
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
    const float m_new = fmaxf(m_run, tile_max);
    const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
    float rsum = 0.f;
#pragma unroll
    for (int i = 0; i < 32; ++i) {
      // exp(-inf - m_new) underflows to 0 — masked slots need no select
      const float p = __expf(vals[i] - m_new);
      vals[i] = p;
      rsum += p;
    }
    rsum += __shfl_xor(rsum, 32, 64);
    l_run = l_run * alpha + rsum;
    m_run = m_new;
#pragma unroll
    for (int db = 0; db < 4; ++db)
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;

    // ---- P^T -> bf16 PV B-fragments via cvt_pk + permlane32_swap (T12) ----
    // per 16-kv group g (4 groups over kv64): B-frag = 4 u32 words where
    // lane needs P^T[kv = g*16 + lhi*8 + e][q_lane], e = 0..7.
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      unsigned w[8];
      // group A: kv [sub*32, sub*32+16): vals idx sub*16 + 0..7
      // group B: kv [sub*32+16, +32):    vals idx sub*16 + 8..15
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        const int base = sub * 16 + g * 8;
        unsigned x0 = cvt_pk_bf16(vals[base + 0], vals[base + 1]);  // kv(0,1)+4*lhi
        unsigned y0 = cvt_pk_bf16(vals[base + 4], vals[base + 5]);  // kv(8,9)+4*lhi
        unsigned x1 = cvt_pk_bf16(vals[base + 2], vals[base + 3]);  // kv(2,3)+4*lhi
        unsigned y1 = cvt_pk_bf16(vals[base + 6], vals[base + 7]);  // kv(10,11)+4*lhi
        auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
        // after swap: lo lane r0 = (kv01, kv45), r1 = (kv23, kv67)
        //             hi lane r0 = (kv89, kv12-13), r1 = (kv10-11, kv14-15)
        w[g * 4 + 0] = (unsigned)r0[0];
        w[g * 4 + 1] = (unsigned)r1[0];
        w[g * 4 + 2] = (unsigned)r0[1];
        w[g * 4 + 3] = (unsigned)r1[1];
      }
      // ---- PV: O^T[d][q] += V^T[d][kv16] @ P^T[kv16][q] for the 2 groups ----
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        short8 pfrag;
        unsigned* pw = reinterpret_cast<unsigned*>(&pfrag);
        pw[0] = w[g * 4 + 0];
        pw[1] = w[g * 4 + 1];
        pw[2] = w[g * 4 + 2];
        pw[3] = w[g * 4 + 3];
        const int kvg = sub * 32 + g * 16;  // kv chunk base within tile
#pragma unroll
        for (int db = 0; db < 4; ++db) {
          // A = V^T rows: lane reads VT[db*32 + l31][kvg + lhi*8 ..+7]
          short8 vf = v2_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);
          o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();  // tile consumed; safe to restage
  }

  // ---- epilogue: O^T /= l, coalesced store into [D][S] ----
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
  for (int db = 0; db < 4; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
      OTh[(long long)d * S + q_lane] = f2bf(o_acc[db][r] * inv_l);
    }
}

// ---------------------------------------------------------------------------
// v3: same per-wave structure as v2, but EIGHT waves per block (512 threads,
// 256 q rows) sharing one staged K/V tile — staging bytes and barrier count
// per flop halve at unchanged per-wave VGPR (124 -> still 4 waves/SIMD via
// 2 blocks/CU).  S % 128 == 0; q rows past S are inert (guarded load/store).
// ---------------------------------------------------------------------------
#define V3_QBLK 256

__device__ __forceinline__ void v3_stage(const ushort* __restrict__ src,
                                         long long ld, int chunks_per_row,
                                         ushort* lds_tile, int chunks_total,
                                         int tid) {
  const int cmask = chunks_per_row - 1;
  const int wave_chunk = tid & ~63;
  for (int s0 = 0; s0 < chunks_total; s0 += 512) {
    const int s = s0 + tid;
    if (s >= chunks_total) break;
    const int row = s / chunks_per_row;
    const int cl = s % chunks_per_row;
    const int c = (cl ^ row) & cmask;
    const ushort* g = src + (long long)row * ld + c * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_tile + ((long long)(s0 + wave_chunk)) * 8),
        16, 0, 0);
  }
}

extern "C" __global__ void __launch_bounds__(512)
attn_fwd_v3_kernel(const ushort* __restrict__ Q, const ushort* __restrict__ K,
                   const ushort* __restrict__ VT, ushort* __restrict__ OT,
                   int B, int H, int Hk, int S, float scale) {
  const int qb = gridDim.x - 1 - blockIdx.x;  // heaviest-first
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (H / Hk);
  const int D = 128;

  const ushort* Qh = Q + (((long long)b * H + h) * S) * D;
  const ushort* Kh = K + (((long long)b * Hk + kvh) * S) * D;
  const ushort* VTh = VT + (((long long)b * Hk + kvh) * D) * S;
  ushort* OTh = OT + (((long long)b * H + h) * D) * S;

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int lhi = lane >> 5;
  const int q0 = qb * V3_QBLK + wid * 32;
  const int q_lane = q0 + l31;
  const bool live = q_lane < S;

  // double-buffered K/V tiles: 2 x 32 KiB — at 2 blocks/CU (VGPR-limited)
  // this still fits 160 KiB LDS, so the prefetch of tile t+1 overlaps the
  // MFMA work of tile t for free (v2/v3 single-buffer relied on occupancy
  // alone; PMC showed 3.99e9 parked cycles at the bench shape)
  __shared__ __attribute__((aligned(16))) ushort smem[2][V2_KVBLK * 128 + 128 * V2_KVBLK];

  short8 qf[8];
  {
    const long long qrow = (long long)(live ? q_lane : 0) * D;
#pragma unroll
    for (int st = 0; st < 8; ++st)
      qf[st] = *reinterpret_cast<const short8*>(Qh + qrow + st * 16 + lhi * 8);
  }

  float m_run = -INFINITY, l_run = 0.f;
  f32x16 o_acc[4];
#pragma unroll
  for (int db = 0; db < 4; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[db][r] = 0.f;

  const int kv_end = min(S, qb * V3_QBLK + V3_QBLK);
  v3_stage(Kh, D, 16, smem[0], V2_KVBLK * 16, tid);
  v3_stage(VTh, S, 8, smem[0] + V2_KVBLK * 128, 128 * 8, tid);
  __syncthreads();
  int buf = 0;
  for (int kv0 = 0; kv0 < kv_end; kv0 += V2_KVBLK) {
    if (kv0 + V2_KVBLK < kv_end) {
      v3_stage(Kh + (long long)(kv0 + V2_KVBLK) * D, D, 16, smem[buf ^ 1],
               V2_KVBLK * 16, tid);
      v3_stage(VTh + kv0 + V2_KVBLK, S, 8, smem[buf ^ 1] + V2_KVBLK * 128,
               128 * 8, tid);
    }
    const ushort* k_lds = smem[buf];
    const ushort* vt_lds = smem[buf] + V2_KVBLK * 128;

    __builtin_amdgcn_s_setprio(1);
    f32x16 st[2];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
      for (int stp = 0; stp < 8; ++stp) {
        short8 kf = v2_read(k_lds, sub * 32 + l31, stp * 2 + lhi, 16);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], acc, 0, 0, 0);
      }
      st[sub] = acc;
    }

    float vals[32];
    float tile_max = -INFINITY;
#pragma unroll
    for (int sub = 0; sub < 2; ++sub)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
        float v = st[sub][r] * scale;
        if (kv > q_lane) v = -INFINITY;
        vals[sub * 16 + r] = v;
        tile_max = fmaxf(tile_max, v);
      }
    __builtin_amdgcn_s_setprio(0);
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
    const float m_new = fmaxf(m_run, tile_max);
    const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
    float rsum = 0.f;
#pragma unroll
    for (int i = 0; i < 32; ++i) {
      const float p = __expf(vals[i] - m_new);
      vals[i] = p;
      rsum += p;
    }
    rsum += __shfl_xor(rsum, 32, 64);
    l_run = l_run * alpha + rsum;
    m_run = m_new;
#pragma unroll
    for (int db = 0; db < 4; ++db)
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;

#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      unsigned w[8];
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        const int base = sub * 16 + g * 8;
        unsigned x0 = cvt_pk_bf16(vals[base + 0], vals[base + 1]);
        unsigned y0 = cvt_pk_bf16(vals[base + 4], vals[base + 5]);
        unsigned x1 = cvt_pk_bf16(vals[base + 2], vals[base + 3]);
        unsigned y1 = cvt_pk_bf16(vals[base + 6], vals[base + 7]);
        auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
        w[g * 4 + 0] = (unsigned)r0[0];
        w[g * 4 + 1] = (unsigned)r1[0];
        w[g * 4 + 2] = (unsigned)r0[1];
        w[g * 4 + 3] = (unsigned)r1[1];
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        short8 pfrag;
        unsigned* pw = reinterpret_cast<unsigned*>(&pfrag);
        pw[0] = w[g * 4 + 0];
        pw[1] = w[g * 4 + 1];
        pw[2] = w[g * 4 + 2];
        pw[3] = w[g * 4 + 3];
        const int kvg = sub * 32 + g * 16;
#pragma unroll
        for (int db = 0; db < 4; ++db) {
          short8 vf = v2_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);
          o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    buf ^= 1;
  }

  if (!live) return;
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
  for (int db = 0; db < 4; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
      OTh[(long long)d * S + q_lane] = f2bf(o_acc[db][r] * inv_l);
    }
}
