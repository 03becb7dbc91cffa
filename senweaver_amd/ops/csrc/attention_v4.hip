// Flash attention forward v4 probe family (gfx950): the v3 8-wave swapped
// QK^T structure with the guide's remaining ladder levers, each selectable
// so a single interleaved probe run prices them (methodology rule 24):
//
//   STAGE modes (var >> 1):
//     0  glds prefetch issued BEFORE QK (v3 as shipped) — anchor
//     1  reg-stage, SINGLE LDS buffer: global_loads issued after QK,
//        vmcnt+LDS-write after a post-PV barrier (the ladder's
//        "async-STAGE split", +17% there; 2 barriers/tile)
//     2  reg-stage, DOUBLE LDS buffer: loads after QK, write into the
//        idle buffer before PV (1 barrier/tile, short register lifetime)
//     3  glds prefetch issued AFTER QK (v3 minus the stage-first
//        anti-pattern that cost 5-15% on the GEMM pipeline)
//   var & 1: defer-max RESCALE_THRESHOLD=8 — skip the O-rescale while the
//        tile max stays within 8 of the running max (P bounded by e^8).
//
// The winning variant is promoted to the attn_fwd_t dispatch; losers stay
// documented in profiles/.  Reference parity: reference flash attention
// behavior; numerics asserted against the fp32 torch
// reference in tests/test_kernels_gpu.py.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define V4_KVBLK 64
#define V4_QBLK 256

static __device__ __forceinline__ void v4_stage_glds(
    const ushort* __restrict__ src, long long ld, int chunks_per_row,
    ushort* lds_tile, int chunks_total, int tid, int nt = 512) {
  const int cmask = chunks_per_row - 1;
  const int wave_chunk = tid & ~63;
  for (int s0 = 0; s0 < chunks_total; s0 += nt) {
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

static __device__ __forceinline__ short8 v4_read(const ushort* lds_tile,
                                                 int row, int c,
                                                 int chunks_per_row) {
  const int phys = (c ^ row) & (chunks_per_row - 1);
  return *reinterpret_cast<const short8*>(lds_tile + (row * chunks_per_row + phys) * 8);
}

// register-staging: the same source chunk the glds path fetches for logical
// slot s, loaded to a VGPR pair instead (16 B)
static __device__ __forceinline__ short8 v4_gload(const ushort* __restrict__ src,
                                                  long long ld,
                                                  int chunks_per_row, int s) {
  const int row = s / chunks_per_row;
  const int cl = s % chunks_per_row;
  const int c = (cl ^ row) & (chunks_per_row - 1);
  return *reinterpret_cast<const short8*>(src + (long long)row * ld + c * 8);
}

static __device__ __forceinline__ unsigned v4_cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

static __device__ __forceinline__ float v4_max(float a, float b) {
  // plain v_max_f32: fmaxf on MFMA outputs gets a canonicalising v_max
  // prepended at -O3 (guide §6 note) — 2 VALU where 1 does
  float r;
  asm("v_max_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

static __device__ __forceinline__ float v4_exp2(float x) {
  float r;
  asm("v_exp_f32 %0, %1" : "=v"(r) : "v"(x));
  return r;
}

template <int STAGE, int DEFER, int DIET = 0, int SPLIT = 0,
          int NT = 512, int TEPI = 0>
static __device__ __forceinline__ void attn4_body(
    const ushort* __restrict__ Q, const ushort* __restrict__ K,
    const ushort* __restrict__ VT, ushort* __restrict__ OT, int B, int H,
    int Hk, int S, float scale) {
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
  constexpr int QBLK = NT / 2;  // 32 q rows per wave
  const int q0 = qb * QBLK + wid * 32;
  const int q_lane = q0 + l31;
  const bool live = q_lane < S;

  // single buffer for STAGE 1; double for 0/2/3
  constexpr int NBUF = (STAGE == 1) ? 1 : 2;
  __shared__ __attribute__((aligned(16))) ushort smem[NBUF][V4_KVBLK * 128 + 128 * V4_KVBLK];

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

  const int kv_end = min(S, qb * QBLK + QBLK);
  v4_stage_glds(Kh, D, 16, smem[0], V4_KVBLK * 16, tid, NT);
  v4_stage_glds(VTh, S, 8, smem[0] + V4_KVBLK * 128, 128 * 8, tid, NT);
  __syncthreads();
  int buf = 0;
  for (int kv0 = 0; kv0 < kv_end; kv0 += V4_KVBLK) {
    const bool more = kv0 + V4_KVBLK < kv_end;
    if (STAGE == 0 && more) {  // v3: prefetch issued ahead of the QK reads
      v4_stage_glds(Kh + (long long)(kv0 + V4_KVBLK) * D, D, 16,
                    smem[buf ^ 1], V4_KVBLK * 16, tid, NT);
      v4_stage_glds(VTh + kv0 + V4_KVBLK, S, 8,
                    smem[buf ^ 1] + V4_KVBLK * 128, 128 * 8, tid, NT);
    }
    const ushort* k_lds = smem[buf];
    const ushort* vt_lds = smem[buf] + V4_KVBLK * 128;

    __builtin_amdgcn_s_setprio(1);
    f32x16 st[2];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
      for (int stp = 0; stp < 8; ++stp) {
        short8 kf = v4_read(k_lds, sub * 32 + l31, stp * 2 + lhi, 16);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], acc, 0, 0, 0);
      }
      st[sub] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // stage issue point for the after-QK modes (K rows of tile t are
    // consumed; VT is still live until PV)
    constexpr int KC = 1024 / NT;  // per-thread chunks of K (and of VT)
    short8 stg[2 * KC];
    if ((STAGE == 1 || STAGE == 2) && more) {
      const ushort* Kn = Kh + (long long)(kv0 + V4_KVBLK) * D;
      const ushort* VTn = VTh + kv0 + V4_KVBLK;
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        stg[c] = v4_gload(Kn, D, 16, tid + c * NT);
        stg[KC + c] = v4_gload(VTn, S, 8, tid + c * NT);
      }
    } else if (STAGE == 3 && more) {
      v4_stage_glds(Kh + (long long)(kv0 + V4_KVBLK) * D, D, 16,
                    smem[buf ^ 1], V4_KVBLK * 16, tid, NT);
      v4_stage_glds(VTh + kv0 + V4_KVBLK, S, 8,
                    smem[buf ^ 1] + V4_KVBLK * 128, 128 * 8, tid, NT);
    }

    float vals[32];
    float tile_max = -INFINITY;
    const float scl = DIET ? scale * 1.44269504f : scale;  // exp2 domain
    const bool interior = DIET && (kv0 + V4_KVBLK - 1 <= q0);
    if (interior) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float v = st[sub][r] * scl;
          vals[sub * 16 + r] = v;
          tile_max = v4_max(tile_max, v);
        }
    } else {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
          float v = st[sub][r] * scl;
          if (kv > q_lane) v = -INFINITY;
          vals[sub * 16 + r] = v;
          tile_max = DIET ? v4_max(tile_max, v) : fmaxf(tile_max, v);
        }
    }
    tile_max = DIET ? v4_max(tile_max, __shfl_xor(tile_max, 32, 64))
                    : fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
    float m_new, alpha;
    bool defer = false;
    if (DEFER) {
      // wave-uniform: every lane's tile max within THR of its running max
      // (first tile: m_run = -inf makes the difference +inf -> no defer)
      defer = __all((int)(tile_max - m_run <= (DIET ? 11.54f : 8.f)));
    }
    if (defer) {
      m_new = m_run;
      alpha = 1.f;
    } else {
      m_new = fmaxf(m_run, tile_max);
      alpha = (m_run == -INFINITY)
                  ? 0.f
                  : (DIET ? v4_exp2(m_run - m_new) : __expf(m_run - m_new));
    }
    float rsum = 0.f;
    if (!SPLIT) {
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        const float p = DIET ? v4_exp2(vals[i] - m_new) : __expf(vals[i] - m_new);
        vals[i] = p;
        rsum += p;
      }
    }
    m_run = m_new;
    if (!defer) {
#pragma unroll
      for (int db = 0; db < 4; ++db)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;
    }

    // STAGE 2: park the fetched tile in the idle buffer before PV (waits
    // the loads here; QK+softmax has been covering the HBM latency)
    if (STAGE == 2 && more) {
      ushort* kd = smem[buf ^ 1];
      ushort* vd = smem[buf ^ 1] + V4_KVBLK * 128;
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        *reinterpret_cast<short8*>(kd + (long long)(tid + c * NT) * 8) = stg[c];
        *reinterpret_cast<short8*>(vd + (long long)(tid + c * NT) * 8) = stg[KC + c];
      }
    }

    if (SPLIT) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int g = 0; g < 2; ++g) {
          const int base = sub * 16 + g * 8;
          float p[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            p[i] = DIET ? v4_exp2(vals[base + i] - m_new)
                        : __expf(vals[base + i] - m_new);
            rsum += p[i];
          }
          unsigned x0 = v4_cvt_pk_bf16(p[0], p[1]);
          unsigned y0 = v4_cvt_pk_bf16(p[4], p[5]);
          unsigned x1 = v4_cvt_pk_bf16(p[2], p[3]);
          unsigned y1 = v4_cvt_pk_bf16(p[6], p[7]);
          auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
          auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
          short8 pfrag;
          unsigned* pw = reinterpret_cast<unsigned*>(&pfrag);
          pw[0] = (unsigned)r0[0];
          pw[1] = (unsigned)r1[0];
          pw[2] = (unsigned)r0[1];
          pw[3] = (unsigned)r1[1];
          const int kvg = sub * 32 + g * 16;
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int db = 0; db < 4; ++db) {
            short8 vf = v4_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);
            o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0);
          }
          __builtin_amdgcn_s_setprio(0);
        }
      }
    } else {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        unsigned w[8];
  #pragma unroll
        for (int g = 0; g < 2; ++g) {
          const int base = sub * 16 + g * 8;
          unsigned x0 = v4_cvt_pk_bf16(vals[base + 0], vals[base + 1]);
          unsigned y0 = v4_cvt_pk_bf16(vals[base + 4], vals[base + 5]);
          unsigned x1 = v4_cvt_pk_bf16(vals[base + 2], vals[base + 3]);
          unsigned y1 = v4_cvt_pk_bf16(vals[base + 6], vals[base + 7]);
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
            short8 vf = v4_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);
            o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
      }
    }
    rsum += __shfl_xor(rsum, 32, 64);
    l_run = l_run * alpha + rsum;

    if (STAGE == 1) {
      __syncthreads();  // every wave done READING the single buffer
      if (more) {
        ushort* kd = smem[0];
        ushort* vd = smem[0] + V4_KVBLK * 128;
#pragma unroll
        for (int c = 0; c < KC; ++c) {
          *reinterpret_cast<short8*>(kd + (long long)(tid + c * NT) * 8) = stg[c];
          *reinterpret_cast<short8*>(vd + (long long)(tid + c * NT) * 8) = stg[KC + c];
        }
      }
      __syncthreads();  // writes visible
    } else {
      __syncthreads();
      buf ^= 1;
    }
  }

  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (TEPI) {
    // T21: per-lane column stores are store-ISSUE-bound (64 b16 stores).
    // Park O in LDS [d][q-in-block] and store whole rows as b128 —
    // QBLK*16/NT (=8) store issues per thread instead of 64.
    __syncthreads();  // last tile's K/V reads done; smem[0] is free
    ushort* o_lds = smem[0];  // [128][QBLK] bf16 = QBLK*256 bytes <= 64 KiB
    const int qib = wid * 32 + l31;
#pragma unroll
    for (int db = 0; db < 4; ++db)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
        o_lds[d * QBLK + qib] = f2bf(o_acc[db][r] * inv_l);
      }
    __syncthreads();
    const int qbase = qb * QBLK;
    constexpr int NCH = 16 * QBLK / NT;  // b128 chunks per thread
#pragma unroll
    for (int c = 0; c < NCH; ++c) {
      const int ch = tid + c * NT;           // chunk over [128][QBLK/8]
      const int d = ch / (QBLK / 8);
      const int qc = (ch % (QBLK / 8)) * 8;  // q offset within block
      if (qbase + qc < S)
        *reinterpret_cast<short8*>(OTh + (long long)d * S + qbase + qc) =
            *reinterpret_cast<const short8*>(o_lds + d * QBLK + qc);
    }
    return;
  }
  if (!live) return;
#pragma unroll
  for (int db = 0; db < 4; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
      OTh[(long long)d * S + q_lane] = f2bf(o_acc[db][r] * inv_l);
    }
}

// ---------------------------------------------------------------------------
// 2-tiles-ahead register staging (PMC: v5's WAIT_ANY ~485 cyc/tile-wave ==
// the exposed tail of HBM latency under a 1-tile-ahead prefetch).  Loads for
// tile t+2 issue after QK(t); the regs parked before PV were issued a full
// tile earlier, so their vmcnt wait is free.  Ping-pong register banks via a
// 2x-unrolled loop (a runtime-indexed bank would go to scratch, rule 20).
// DIET adds the softmax VALU cuts: exp2-domain fold (v_exp_f32 IS exp2 —
// the libm mul by log2e folds into the scale), raw v_max_f32, and the
// interior-tile mask skip (tiles fully below the diagonal never mask).
// ---------------------------------------------------------------------------
template <int DEFER, int DIET>
static __device__ __forceinline__ void attn4_body2(
    const ushort* __restrict__ Q, const ushort* __restrict__ K,
    const ushort* __restrict__ VT, ushort* __restrict__ OT, int B, int H,
    int Hk, int S, float scale) {
  const int qb = gridDim.x - 1 - blockIdx.x;
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
  const int q0 = qb * V4_QBLK + wid * 32;
  const int q_lane = q0 + l31;
  const bool live = q_lane < S;

  __shared__ __attribute__((aligned(16))) ushort smem[2][V4_KVBLK * 128 + 128 * V4_KVBLK];

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

  const int kv_end = min(S, qb * V4_QBLK + V4_QBLK);
  const float scl = DIET ? scale * 1.44269504f : scale;  // exp2 domain

  // prologue: tile 0 -> LDS[0]; tile 1 -> register bank A
  v4_stage_glds(Kh, D, 16, smem[0], V4_KVBLK * 16, tid);
  v4_stage_glds(VTh, S, 8, smem[0] + V4_KVBLK * 128, 128 * 8, tid);
  short8 stgA[4], stgB[4];
  if (V4_KVBLK < kv_end) {
    const ushort* Kn = Kh + (long long)V4_KVBLK * D;
    const ushort* VTn = VTh + V4_KVBLK;
    stgA[0] = v4_gload(Kn, D, 16, tid);
    stgA[1] = v4_gload(Kn, D, 16, tid + 512);
    stgA[2] = v4_gload(VTn, S, 8, tid);
    stgA[3] = v4_gload(VTn, S, 8, tid + 512);
  }
  __syncthreads();

  int buf = 0;
  int kv0 = 0;
#define V4B2_TILE(CUR, NXT)                                                   \
  {                                                                           \
    const bool more = kv0 + V4_KVBLK < kv_end;                                \
    const bool more2 = kv0 + 2 * V4_KVBLK < kv_end;                           \
    const ushort* k_lds = smem[buf];                                          \
    const ushort* vt_lds = smem[buf] + V4_KVBLK * 128;                        \
    __builtin_amdgcn_s_setprio(1);                                            \
    f32x16 st[2];                                                             \
    _Pragma("unroll") for (int sub = 0; sub < 2; ++sub) {                     \
      f32x16 acc;                                                             \
      _Pragma("unroll") for (int r = 0; r < 16; ++r) acc[r] = 0.f;            \
      _Pragma("unroll") for (int stp = 0; stp < 8; ++stp) {                   \
        short8 kf = v4_read(k_lds, sub * 32 + l31, stp * 2 + lhi, 16);        \
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], acc, 0, 0, 0); \
      }                                                                       \
      st[sub] = acc;                                                          \
    }                                                                         \
    __builtin_amdgcn_s_setprio(0);                                            \
    if (more2) {                                                              \
      const ushort* Kn = Kh + (long long)(kv0 + 2 * V4_KVBLK) * D;            \
      const ushort* VTn = VTh + kv0 + 2 * V4_KVBLK;                           \
      NXT[0] = v4_gload(Kn, D, 16, tid);                                      \
      NXT[1] = v4_gload(Kn, D, 16, tid + 512);                                \
      NXT[2] = v4_gload(VTn, S, 8, tid);                                      \
      NXT[3] = v4_gload(VTn, S, 8, tid + 512);                                \
    }                                                                         \
    float vals[32];                                                           \
    float tile_max = -INFINITY;                                               \
    const bool interior = DIET && (kv0 + V4_KVBLK - 1 <= q0);                 \
    if (interior) {                                                           \
      _Pragma("unroll") for (int sub = 0; sub < 2; ++sub)                     \
      _Pragma("unroll") for (int r = 0; r < 16; ++r) {                        \
        const float v = st[sub][r] * scl;                                     \
        vals[sub * 16 + r] = v;                                               \
        tile_max = v4_max(tile_max, v);                                       \
      }                                                                       \
    } else {                                                                  \
      _Pragma("unroll") for (int sub = 0; sub < 2; ++sub)                     \
      _Pragma("unroll") for (int r = 0; r < 16; ++r) {                        \
        const int kv = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;     \
        float v = st[sub][r] * scl;                                           \
        if (kv > q_lane) v = -INFINITY;                                       \
        vals[sub * 16 + r] = v;                                               \
        tile_max = DIET ? v4_max(tile_max, v) : fmaxf(tile_max, v);           \
      }                                                                       \
    }                                                                         \
    tile_max = DIET ? v4_max(tile_max, __shfl_xor(tile_max, 32, 64))          \
                    : fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));          \
    float m_new, alpha;                                                       \
    bool defer = false;                                                       \
    if (DEFER) defer = __all((int)(tile_max - m_run <= (DIET ? 11.54f : 8.f))); \
    if (defer) {                                                              \
      m_new = m_run;                                                          \
      alpha = 1.f;                                                            \
    } else {                                                                  \
      m_new = fmaxf(m_run, tile_max);                                         \
      alpha = (m_run == -INFINITY)                                            \
                  ? 0.f                                                       \
                  : (DIET ? v4_exp2(m_run - m_new) : __expf(m_run - m_new));  \
    }                                                                         \
    float rsum = 0.f;                                                         \
    _Pragma("unroll") for (int i = 0; i < 32; ++i) {                          \
      const float p = DIET ? v4_exp2(vals[i] - m_new) : __expf(vals[i] - m_new); \
      vals[i] = p;                                                            \
      rsum += p;                                                              \
    }                                                                         \
    rsum += __shfl_xor(rsum, 32, 64);                                         \
    l_run = l_run * alpha + rsum;                                             \
    m_run = m_new;                                                            \
    if (!defer) {                                                             \
      _Pragma("unroll") for (int db = 0; db < 4; ++db)                        \
      _Pragma("unroll") for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;   \
    }                                                                         \
    if (more) { /* park tile t+1 (issued a full tile ago) in the idle buf */  \
      ushort* kd = smem[buf ^ 1];                                             \
      ushort* vd = smem[buf ^ 1] + V4_KVBLK * 128;                            \
      *reinterpret_cast<short8*>(kd + (long long)tid * 8) = CUR[0];           \
      *reinterpret_cast<short8*>(kd + (long long)(tid + 512) * 8) = CUR[1];   \
      *reinterpret_cast<short8*>(vd + (long long)tid * 8) = CUR[2];           \
      *reinterpret_cast<short8*>(vd + (long long)(tid + 512) * 8) = CUR[3];   \
    }                                                                         \
    _Pragma("unroll") for (int sub = 0; sub < 2; ++sub) {                     \
      unsigned w[8];                                                          \
      _Pragma("unroll") for (int g = 0; g < 2; ++g) {                         \
        const int base = sub * 16 + g * 8;                                    \
        unsigned x0 = v4_cvt_pk_bf16(vals[base + 0], vals[base + 1]);         \
        unsigned y0 = v4_cvt_pk_bf16(vals[base + 4], vals[base + 5]);         \
        unsigned x1 = v4_cvt_pk_bf16(vals[base + 2], vals[base + 3]);         \
        unsigned y1 = v4_cvt_pk_bf16(vals[base + 6], vals[base + 7]);         \
        auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);     \
        auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);     \
        w[g * 4 + 0] = (unsigned)r0[0];                                       \
        w[g * 4 + 1] = (unsigned)r1[0];                                       \
        w[g * 4 + 2] = (unsigned)r0[1];                                       \
        w[g * 4 + 3] = (unsigned)r1[1];                                       \
      }                                                                       \
      __builtin_amdgcn_s_setprio(1);                                          \
      _Pragma("unroll") for (int g = 0; g < 2; ++g) {                         \
        short8 pfrag;                                                         \
        unsigned* pw = reinterpret_cast<unsigned*>(&pfrag);                   \
        pw[0] = w[g * 4 + 0];                                                 \
        pw[1] = w[g * 4 + 1];                                                 \
        pw[2] = w[g * 4 + 2];                                                 \
        pw[3] = w[g * 4 + 3];                                                 \
        const int kvg = sub * 32 + g * 16;                                    \
        _Pragma("unroll") for (int db = 0; db < 4; ++db) {                    \
          short8 vf = v4_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);    \
          o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0); \
        }                                                                     \
      }                                                                       \
      __builtin_amdgcn_s_setprio(0);                                          \
    }                                                                         \
    __syncthreads();                                                          \
    buf ^= 1;                                                                 \
    kv0 += V4_KVBLK;                                                          \
  }

  while (kv0 < kv_end) {
    V4B2_TILE(stgA, stgB);
    if (kv0 >= kv_end) break;
    V4B2_TILE(stgB, stgA);
  }
#undef V4B2_TILE

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


// ---------------------------------------------------------------------------
// v6: persistent workgroups + cross-seam prefetch.  At S=2048 the launched
// v15 pays ~4 us per 256-row block (prologue stage stall + Q load + tail)
// and blocks run back-to-back on a CU at 1 block/CU — fully serialized.
// Here gridDim.x (= #CUs' worth) blocks walk work items; during an item's
// LAST tile the regular staging path parks the NEXT item's tile 0 in the
// idle buffer, so the seam costs only a qf reload.  Work items decode with
// qb heaviest-first, matching the launched mapping.
// ---------------------------------------------------------------------------
template <int DEFER, int DIET>
static __device__ __forceinline__ void attn4_body_p(
    const ushort* __restrict__ Q, const ushort* __restrict__ K,
    const ushort* __restrict__ VT, ushort* __restrict__ OT, int B, int H,
    int Hk, int S, float scale) {
  const int D = 128;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int l31 = lane & 31;
  const int lhi = lane >> 5;
  const int nqb = (S + 255) / 256;
  const int NW = nqb * H * B;

  __shared__ __attribute__((aligned(16))) ushort smem[2][V4_KVBLK * 128 + 128 * V4_KVBLK];

  const float scl = DIET ? scale * 1.44269504f : scale;
  int buf = 0;
  bool first = true;

  for (int item = blockIdx.x; item < NW; item += gridDim.x) {
    // decode work item (qb heaviest-first, like the launched grid)
    const int x = item % nqb;
    const int qb = nqb - 1 - x;
    const int h = (item / nqb) % H;
    const int b = item / (nqb * H);
    const int kvh = h / (H / Hk);
    const ushort* Qh = Q + (((long long)b * H + h) * S) * D;
    const ushort* Kh = K + (((long long)b * Hk + kvh) * S) * D;
    const ushort* VTh = VT + (((long long)b * Hk + kvh) * D) * S;
    ushort* OTh = OT + (((long long)b * H + h) * D) * S;
    const int q0 = qb * V4_QBLK + wid * 32;
    const int q_lane = q0 + l31;
    const bool live = q_lane < S;
    const int kv_end = min(S, qb * V4_QBLK + V4_QBLK);

    const bool more_items = item + (int)gridDim.x < NW;

    if (first) {  // only ever once: later items find tile 0 parked
      v4_stage_glds(Kh, D, 16, smem[buf], V4_KVBLK * 16, tid, 512);
      v4_stage_glds(VTh, S, 8, smem[buf] + V4_KVBLK * 128, 128 * 8, tid, 512);
      first = false;
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }

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

    for (int kv0 = 0; kv0 < kv_end; kv0 += V4_KVBLK) {
      const bool more_tiles = kv0 + V4_KVBLK < kv_end;
      const bool park = more_tiles || more_items;
      const ushort* k_lds = smem[buf];
      const ushort* vt_lds = smem[buf] + V4_KVBLK * 128;

      __builtin_amdgcn_s_setprio(1);
      f32x16 st[2];
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        f32x16 acc;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
        for (int stp = 0; stp < 8; ++stp) {
          short8 kf = v4_read(k_lds, sub * 32 + l31, stp * 2 + lhi, 16);
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], acc, 0, 0, 0);
        }
        st[sub] = acc;
      }
      __builtin_amdgcn_s_setprio(0);

      // stage: next tile of THIS item, or tile 0 of the NEXT item
      // (next-item bases decoded HERE so they are never live across the
      // tile loop — computed at item top they spilled 388 B/lane)
      short8 stg[4];
      if (park) {
        const ushort* Kn;
        const ushort* VTn;
        if (more_tiles) {
          Kn = Kh + (long long)(kv0 + V4_KVBLK) * D;
          VTn = VTh + kv0 + V4_KVBLK;
        } else {
          const int item2 = item + gridDim.x;
          const int h2 = (item2 / nqb) % H;
          const int b2 = item2 / (nqb * H);
          const int kvh2 = h2 / (H / Hk);
          Kn = K + (((long long)b2 * Hk + kvh2) * S) * D;
          VTn = VT + (((long long)b2 * Hk + kvh2) * D) * S;
        }
        stg[0] = v4_gload(Kn, D, 16, tid);
        stg[1] = v4_gload(Kn, D, 16, tid + 512);
        stg[2] = v4_gload(VTn, S, 8, tid);
        stg[3] = v4_gload(VTn, S, 8, tid + 512);
      }

      float vals[32];
      float tile_max = -INFINITY;
      const bool interior = DIET && (kv0 + V4_KVBLK - 1 <= q0);
      if (interior) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float v = st[sub][r] * scl;
            vals[sub * 16 + r] = v;
            tile_max = v4_max(tile_max, v);
          }
      } else {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kv = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
            float v = st[sub][r] * scl;
            if (kv > q_lane) v = -INFINITY;
            vals[sub * 16 + r] = v;
            tile_max = DIET ? v4_max(tile_max, v) : fmaxf(tile_max, v);
          }
      }
      tile_max = DIET ? v4_max(tile_max, __shfl_xor(tile_max, 32, 64))
                      : fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
      float m_new, alpha;
      bool defer = false;
      if (DEFER) defer = __all((int)(tile_max - m_run <= (DIET ? 11.54f : 8.f)));
      if (defer) {
        m_new = m_run;
        alpha = 1.f;
      } else {
        m_new = fmaxf(m_run, tile_max);
        alpha = (m_run == -INFINITY)
                    ? 0.f
                    : (DIET ? v4_exp2(m_run - m_new) : __expf(m_run - m_new));
      }
      float rsum = 0.f;
      m_run = m_new;
      if (!defer) {
#pragma unroll
        for (int db = 0; db < 4; ++db)
#pragma unroll
          for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;
      }

      if (park) {
        ushort* kd = smem[buf ^ 1];
        ushort* vd = smem[buf ^ 1] + V4_KVBLK * 128;
        *reinterpret_cast<short8*>(kd + (long long)tid * 8) = stg[0];
        *reinterpret_cast<short8*>(kd + (long long)(tid + 512) * 8) = stg[1];
        *reinterpret_cast<short8*>(vd + (long long)tid * 8) = stg[2];
        *reinterpret_cast<short8*>(vd + (long long)(tid + 512) * 8) = stg[3];
      }

      // sm-split PV
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int g = 0; g < 2; ++g) {
          const int base = sub * 16 + g * 8;
          float p[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            p[i] = DIET ? v4_exp2(vals[base + i] - m_new)
                        : __expf(vals[base + i] - m_new);
            rsum += p[i];
          }
          unsigned x0 = v4_cvt_pk_bf16(p[0], p[1]);
          unsigned y0 = v4_cvt_pk_bf16(p[4], p[5]);
          unsigned x1 = v4_cvt_pk_bf16(p[2], p[3]);
          unsigned y1 = v4_cvt_pk_bf16(p[6], p[7]);
          auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
          auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
          short8 pfrag;
          unsigned* pw = reinterpret_cast<unsigned*>(&pfrag);
          pw[0] = (unsigned)r0[0];
          pw[1] = (unsigned)r1[0];
          pw[2] = (unsigned)r0[1];
          pw[3] = (unsigned)r1[1];
          const int kvg = sub * 32 + g * 16;
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int db = 0; db < 4; ++db) {
            short8 vf = v4_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);
            o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0);
          }
          __builtin_amdgcn_s_setprio(0);
        }
      }
      rsum += __shfl_xor(rsum, 32, 64);
      l_run = l_run * alpha + rsum;

      __syncthreads();
      buf ^= 1;
    }

    // epilogue for this item (no LDS use: safe to overlap other waves'
    // next-item QK against the parked buffer)
    if (live) {
      const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
      for (int db = 0; db < 4; ++db)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
          OTh[(long long)d * S + q_lane] = f2bf(o_acc[db][r] * inv_l);
        }
    }
  }
}

#define V4_KERNEL(VAR, STAGE, DEFER)                                          \
  extern "C" __global__ void __launch_bounds__(512)                           \
      attn_fwd_v4_##VAR##_kernel(const ushort* __restrict__ Q,                \
                                 const ushort* __restrict__ K,                \
                                 const ushort* __restrict__ VT,               \
                                 ushort* __restrict__ OT, int B, int H,       \
                                 int Hk, int S, float scale) {                \
    attn4_body<STAGE, DEFER>(Q, K, VT, OT, B, H, Hk, S, scale);               \
  }

V4_KERNEL(0, 0, 0)  // v3 anchor
V4_KERNEL(1, 0, 1)  // + defer-max
V4_KERNEL(2, 1, 0)  // async-STAGE single-buffer
V4_KERNEL(3, 1, 1)
V4_KERNEL(4, 2, 0)  // reg-stage double-buffer, write pre-PV
V4_KERNEL(5, 2, 1)
V4_KERNEL(6, 3, 0)  // glds after QK
V4_KERNEL(7, 3, 1)

// ---------------------------------------------------------------------------
// 2-tiles-ahead via glds into a 3-slot LDS ring, counted s_waitcnt + raw
// s_barrier (the gemm_pipe v14 idiom): zero register cost — the register-bank
// version (vars 8-10) spills at 256 VGPR.  Per wave per tile: 4 glds issues
// (2 K + 2 VT); boundary waits vmcnt(4) when a prefetch is in flight (the
// next tile's 4 stay outstanding), vmcnt(0) on the tail.  Q-fragment loads
// drain (vmcnt 0) BEFORE the first glds so the manual counts stay exact.
// ---------------------------------------------------------------------------
template <int DEFER, int DIET, int SPLIT = 0>
static __device__ __forceinline__ void attn4_body3(
    const ushort* __restrict__ Q, const ushort* __restrict__ K,
    const ushort* __restrict__ VT, ushort* __restrict__ OT, int B, int H,
    int Hk, int S, float scale) {
  const int qb = gridDim.x - 1 - blockIdx.x;
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
  const int q0 = qb * V4_QBLK + wid * 32;
  const int q_lane = q0 + l31;
  const bool live = q_lane < S;

  constexpr int TILE = V4_KVBLK * 128 + 128 * V4_KVBLK;
  __shared__ __attribute__((aligned(16))) ushort smem[3][TILE];

  short8 qf[8];
  {
    const long long qrow = (long long)(live ? q_lane : 0) * D;
#pragma unroll
    for (int st = 0; st < 8; ++st)
      qf[st] = *reinterpret_cast<const short8*>(Qh + qrow + st * 16 + lhi * 8);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // qf in regs; counts clean

  float m_run = -INFINITY, l_run = 0.f;
  f32x16 o_acc[4];
#pragma unroll
  for (int db = 0; db < 4; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[db][r] = 0.f;

  const int kv_end = min(S, qb * V4_QBLK + V4_QBLK);
  const float scl = DIET ? scale * 1.44269504f : scale;

  v4_stage_glds(Kh, D, 16, smem[0], V4_KVBLK * 16, tid);
  v4_stage_glds(VTh, S, 8, smem[0] + V4_KVBLK * 128, 128 * 8, tid);
  if (V4_KVBLK < kv_end) {
    v4_stage_glds(Kh + (long long)V4_KVBLK * D, D, 16, smem[1],
                  V4_KVBLK * 16, tid);
    v4_stage_glds(VTh + V4_KVBLK, S, 8, smem[1] + V4_KVBLK * 128,
                  128 * 8, tid);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  int slot = 0;
  for (int kv0 = 0; kv0 < kv_end; kv0 += V4_KVBLK) {
    const bool more = kv0 + V4_KVBLK < kv_end;
    const bool more2 = kv0 + 2 * V4_KVBLK < kv_end;
    const ushort* k_lds = smem[slot];
    const ushort* vt_lds = smem[slot] + V4_KVBLK * 128;

    __builtin_amdgcn_s_setprio(1);
    f32x16 st[2];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
      for (int stp = 0; stp < 8; ++stp) {
        short8 kf = v4_read(k_lds, sub * 32 + l31, stp * 2 + lhi, 16);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[stp], acc, 0, 0, 0);
      }
      st[sub] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    if (more2) {  // prefetch t+2 into the slot freed at the last boundary
      const int dst = (slot + 2 >= 3) ? slot - 1 : slot + 2;
      v4_stage_glds(Kh + (long long)(kv0 + 2 * V4_KVBLK) * D, D, 16,
                    smem[dst], V4_KVBLK * 16, tid);
      v4_stage_glds(VTh + kv0 + 2 * V4_KVBLK, S, 8,
                    smem[dst] + V4_KVBLK * 128, 128 * 8, tid);
    }

    float vals[32];
    float tile_max = -INFINITY;
    const bool interior = DIET && (kv0 + V4_KVBLK - 1 <= q0);
    if (interior) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float v = st[sub][r] * scl;
          vals[sub * 16 + r] = v;
          tile_max = v4_max(tile_max, v);
        }
    } else {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
          float v = st[sub][r] * scl;
          if (kv > q_lane) v = -INFINITY;
          vals[sub * 16 + r] = v;
          tile_max = DIET ? v4_max(tile_max, v) : fmaxf(tile_max, v);
        }
    }
    tile_max = DIET ? v4_max(tile_max, __shfl_xor(tile_max, 32, 64))
                    : fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
    float m_new, alpha;
    bool defer = false;
    if (DEFER) defer = __all((int)(tile_max - m_run <= (DIET ? 11.54f : 8.f)));
    if (defer) {
      m_new = m_run;
      alpha = 1.f;
    } else {
      m_new = fmaxf(m_run, tile_max);
      alpha = (m_run == -INFINITY)
                  ? 0.f
                  : (DIET ? v4_exp2(m_run - m_new) : __expf(m_run - m_new));
    }
    float rsum = 0.f;
    if (!SPLIT) {
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        const float p = DIET ? v4_exp2(vals[i] - m_new) : __expf(vals[i] - m_new);
        vals[i] = p;
        rsum += p;
      }
    }
    m_run = m_new;
    if (!defer) {
#pragma unroll
      for (int db = 0; db < 4; ++db)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;
    }

    if (SPLIT) {
      // sm-split: exp of each 8-value group lands right before its PV
      // group, so the trans/VALU chain issues under the previous group's
      // MFMAs instead of as one serial block
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int g = 0; g < 2; ++g) {
          const int base = sub * 16 + g * 8;
          float p[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            p[i] = DIET ? v4_exp2(vals[base + i] - m_new)
                        : __expf(vals[base + i] - m_new);
            rsum += p[i];
          }
          unsigned x0 = v4_cvt_pk_bf16(p[0], p[1]);
          unsigned y0 = v4_cvt_pk_bf16(p[4], p[5]);
          unsigned x1 = v4_cvt_pk_bf16(p[2], p[3]);
          unsigned y1 = v4_cvt_pk_bf16(p[6], p[7]);
          auto r0 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
          auto r1 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
          short8 pfrag;
          unsigned* pw = reinterpret_cast<unsigned*>(&pfrag);
          pw[0] = (unsigned)r0[0];
          pw[1] = (unsigned)r1[0];
          pw[2] = (unsigned)r0[1];
          pw[3] = (unsigned)r1[1];
          const int kvg = sub * 32 + g * 16;
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int db = 0; db < 4; ++db) {
            short8 vf = v4_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);
            o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0);
          }
          __builtin_amdgcn_s_setprio(0);
        }
      }
    } else {
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      unsigned w[8];
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        const int base = sub * 16 + g * 8;
        unsigned x0 = v4_cvt_pk_bf16(vals[base + 0], vals[base + 1]);
        unsigned y0 = v4_cvt_pk_bf16(vals[base + 4], vals[base + 5]);
        unsigned x1 = v4_cvt_pk_bf16(vals[base + 2], vals[base + 3]);
        unsigned y1 = v4_cvt_pk_bf16(vals[base + 6], vals[base + 7]);
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
          short8 vf = v4_read(vt_lds, db * 32 + l31, (kvg >> 3) + lhi, 8);
          o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag, o_acc[db], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    }
    rsum += __shfl_xor(rsum, 32, 64);
    l_run = l_run * alpha + rsum;

    if (more) {
      // MY t+1 glds must have landed before anyone reads the next slot;
      // per-wave wait FIRST, then the rendezvous (vmcnt is per-wave)
      if (more2)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    slot = (slot + 1 >= 3) ? 0 : slot + 1;
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

#define V4_KERNEL2(VAR, DEFER, DIET)                                          \
  extern "C" __global__ void __launch_bounds__(512)                           \
      attn_fwd_v4_##VAR##_kernel(const ushort* __restrict__ Q,                \
                                 const ushort* __restrict__ K,                \
                                 const ushort* __restrict__ VT,               \
                                 ushort* __restrict__ OT, int B, int H,       \
                                 int Hk, int S, float scale) {                \
    attn4_body2<DEFER, DIET>(Q, K, VT, OT, B, H, Hk, S, scale);               \
  }

V4_KERNEL2(8, 0, 0)   // 2-tiles-ahead register staging
V4_KERNEL2(9, 1, 0)   // + defer-max
V4_KERNEL2(10, 1, 1)  // + softmax VALU diet

#define V4_KERNEL3(VAR, DEFER, DIET)                                          \
  extern "C" __global__ void __launch_bounds__(512)                           \
      attn_fwd_v4_##VAR##_kernel(const ushort* __restrict__ Q,                \
                                 const ushort* __restrict__ K,                \
                                 const ushort* __restrict__ VT,               \
                                 ushort* __restrict__ OT, int B, int H,       \
                                 int Hk, int S, float scale) {                \
    attn4_body3<DEFER, DIET>(Q, K, VT, OT, B, H, Hk, S, scale);               \
  }

V4_KERNEL3(11, 1, 0)  // glds 3-ring 2-ahead + defer
V4_KERNEL3(12, 1, 1)  // + softmax VALU diet

extern "C" __global__ void __launch_bounds__(512)
attn_fwd_v4_13_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ VT,
                      ushort* __restrict__ OT, int B, int H, int Hk, int S,
                      float scale) {
  // reg-stage double-buffer (v5) + defer + diet
  attn4_body<2, 1, 1>(Q, K, VT, OT, B, H, Hk, S, scale);
}

extern "C" __global__ void __launch_bounds__(512)
attn_fwd_v4_14_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ VT,
                      ushort* __restrict__ OT, int B, int H, int Hk, int S,
                      float scale) {
  // glds 3-ring + defer + diet + sm-split
  attn4_body3<1, 1, 1>(Q, K, VT, OT, B, H, Hk, S, scale);
}

extern "C" __global__ void __launch_bounds__(512)
attn_fwd_v4_15_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ VT,
                      ushort* __restrict__ OT, int B, int H, int Hk, int S,
                      float scale) {
  // reg-stage double-buffer + defer + diet + sm-split
  attn4_body<2, 1, 1, 1>(Q, K, VT, OT, B, H, Hk, S, scale);
}

extern "C" __global__ void __launch_bounds__(256)
attn_fwd_v4_16_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ VT,
                      ushort* __restrict__ OT, int B, int H, int Hk, int S,
                      float scale) {
  // 4-wave blocks (QBLK 128): two co-resident blocks per CU
  attn4_body<2, 1, 1, 1, 256>(Q, K, VT, OT, B, H, Hk, S, scale);
}

extern "C" __global__ void __launch_bounds__(512)
attn_fwd_v4_17_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ VT,
                      ushort* __restrict__ OT, int B, int H, int Hk, int S,
                      float scale) {
  // v15 + T21 LDS-transposed epilogue
  attn4_body<2, 1, 1, 1, 512, 1>(Q, K, VT, OT, B, H, Hk, S, scale);
}

extern "C" __global__ void __launch_bounds__(256)
attn_fwd_v4_18_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ VT,
                      ushort* __restrict__ OT, int B, int H, int Hk, int S,
                      float scale) {
  // 4-wave blocks + T21 epilogue
  attn4_body<2, 1, 1, 1, 256, 1>(Q, K, VT, OT, B, H, Hk, S, scale);
}

extern "C" __global__ void __launch_bounds__(512)
attn_fwd_v4_19_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ VT,
                      ushort* __restrict__ OT, int B, int H, int Hk, int S,
                      float scale) {
  // persistent + cross-seam prefetch + defer + diet (+ sm-split)
  attn4_body_p<1, 1>(Q, K, VT, OT, B, H, Hk, S, scale);
}
