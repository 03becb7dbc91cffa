#include "hip/hip_runtime.h"
// Flash-style causal attention forward (prefill) for gfx950, D=128, GQA.
//
// The beam scorer is prefill-dominated (teacher-forced log-prob of rollout
// tokens), so this kernel targets throughput at S ~ 1-8k:
//   - grid (S/64, H, B); 256-thread block = 4 waves; each wave owns 16 Q rows
//   - QK^T on v_mfma_f32_16x16x32_bf16 (K stored [S,D] row-major = B^T form,
//     fragments are 16-byte row loads from an XOR-swizzled LDS tile)
//   - online softmax per 16-row group held in registers (C-fragment layout:
//     row=(lane>>4)*4+e, col=lane&15), running (m, l) per row
//   - PV via a wave-private LDS P tile and a pre-transposed V (VT[B,Hk,D,S],
//     transposed once per forward, amortized over H_q/H_kv heads and S/64
//     tiles) so both PV fragments are 16-byte LDS row reads
//   - KV tiles (64x128) staged by global_load_lds_dwordx4 with the same
//     source-side XOR swizzle as the GEMM (rule: swizzle source + read, never
//     the lane-linear glds destination)
//
// f32 softmax/accumulation; bf16 I/O.  S % 64 == 0 (host pads with -inf-
// masked rows via the causal mask: rows >= S_real are never read back).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define KVBLK 64
#define QBLK 64  // per block; 16 per wave

// stage a [rows x row_bf16] tile (row_bf16 = 128 or 64) into lane-linear LDS
// with chunk ^= row&7 source swizzle.  chunks_total = rows * row_bf16 / 8.
__device__ __forceinline__ void stage_swz(const ushort* __restrict__ src,
                                          long long ld, int chunks_per_row,
                                          ushort* lds_tile, int chunks_total,
                                          int tid) {
  // glds lane destination = wave-uniform base + lane*16: the base carries
  // the wave's 64-chunk sub-block.
  const int wave_chunk = tid & ~63;
  for (int s0 = 0; s0 < chunks_total; s0 += 256) {
    const int s = s0 + tid;
    const int row = s / chunks_per_row;
    const int cl = s % chunks_per_row;
    const int c = ((cl & 7) ^ (row & 7)) | (cl & ~7);
    const ushort* g = src + (long long)row * ld + c * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_tile + ((long long)(s0 + wave_chunk)) * 8),
        16, 0, 0);
  }
}

__device__ __forceinline__ short8 read_swz(const ushort* lds_tile, int row,
                                           int c, int chunks_per_row) {
  const int phys = ((c & 7) ^ (row & 7)) | (c & ~7);
  return *reinterpret_cast<const short8*>(lds_tile + (row * chunks_per_row + phys) * 8);
}

extern "C" __global__ void __launch_bounds__(256, 2)
attn_fwd_bf16_kernel(const ushort* __restrict__ Q, const ushort* __restrict__ K,
                     const ushort* __restrict__ VT, ushort* __restrict__ O,
                     int B, int H, int Hk, int S, float scale) {
  // heaviest-first dispatch: causal work grows with the Q block index
  // (block qb touches qb+1 KV tiles), so reverse the launch order to avoid
  // a heavy straggler tail in the final dispatch rounds
  const int qb = gridDim.x - 1 - blockIdx.x;  // Q block of 64 rows
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (H / Hk);
  const int D = 128;

  const ushort* Qh = Q + (((long long)b * H + h) * S) * D;
  const ushort* Kh = K + (((long long)b * Hk + kvh) * S) * D;
  const ushort* VTh = VT + (((long long)b * Hk + kvh) * D) * S;  // [D][S]
  ushort* Oh = O + (((long long)b * H + h) * S) * D;

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int q0 = qb * QBLK + wid * 16;  // this wave's first Q row

  // double-buffered K/VT tiles (32 KiB per buffer) + wave-private P tiles:
  // ONE __shared__ object (multiple objects make hipcc drain vmcnt before
  // every ds_read of a glds pipeline — CDNA4 guide trap (a))
  __shared__ __attribute__((aligned(16))) ushort smem[2 * 2 * KVBLK * 128 + 4 * 16 * KVBLK];
  // layout: [buf][K | VT] tiles then the per-wave P region; buffer bases are
  // computed by index (an initialized pointer array of LDS addresses fails
  // to compile: addrspacecast in static initializer)
  ushort* p_lds = smem + 4 * KVBLK * 128;  // [wave][16 * KVBLK]

  // Q fragments: lane holds Q[q0 + lane%16][kk*32 + (lane>>4)*8 .. +7]
  short8 qf[4];
  {
    const int qrow = q0 + (lane & 15);
    const int kg = lane >> 4;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      qf[kk] = *reinterpret_cast<const short8*>(Qh + (long long)qrow * D + kk * 32 + kg * 8);
  }

  // online-softmax state per q row (4 rows per lane via C-fragment e index)
  float m_run[4], l_run[4];
  f32x4 o_acc[8];  // O[16,128]: 8 n-tiles
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    m_run[e] = -INFINITY;
    l_run[e] = 0.f;
  }
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) o_acc[nt] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = min(S, qb * QBLK + QBLK);  // causal upper bound
  // prologue: stage tile 0 into buffer 0
  stage_swz(Kh, D, 16, smem, KVBLK * 16, tid);
  stage_swz(VTh, S, 8, smem + KVBLK * 128, 128 * 8, tid);
  __syncthreads();
  int cur = 0;
  for (int kv0 = 0; kv0 < kv_end; kv0 += KVBLK) {
    // Pipelined staging with COUNTED vmcnt across raw barriers: the next
    // tile's 8 glds stay in flight through the whole compute phase
    // (PMC showed 55% of wave-cycles parked on the __syncthreads drain).
    // vmcnt(8) = previous tile landed, the 8 just issued may still fly.
    if (kv0 + KVBLK < kv_end) {
      ushort* nk = smem + (cur ^ 1) * (2 * KVBLK * 128);
      stage_swz(Kh + (long long)(kv0 + KVBLK) * D, D, 16, nk, KVBLK * 16, tid);
      stage_swz(VTh + kv0 + KVBLK, S, 8, nk + KVBLK * 128, 128 * 8, tid);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    const ushort* k_lds = smem + cur * (2 * KVBLK * 128);
    const ushort* vt_lds = k_lds + KVBLK * 128;

    // ---- QK^T: S_tile[16 q][64 kv] = Q[16,128] @ K_tile^T ----
    f32x4 s_frag[4];
    const int kg = lane >> 4;
#pragma unroll
    for (int kt = 0; kt < 4; ++kt) {  // 4 kv 16-col tiles
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {  // K=128 in 4 steps of 32
        short8 kfrag = read_swz(k_lds, kt * 16 + (lane & 15), kk * 4 + kg, 16);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[kk], kfrag, acc, 0, 0, 0);
      }
      s_frag[kt] = acc;
    }

    // ---- online softmax over the 64 new columns ----
    const int col_base = kv0 + (lane & 15);
    float p[4][4];  // [kt][e]
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const int qrow = q0 + (lane >> 4) * 4 + e;
      float rmax = -INFINITY;
#pragma unroll
      for (int kt = 0; kt < 4; ++kt) {
        float sv = s_frag[kt][e] * scale;
        if (col_base + kt * 16 > qrow) sv = -INFINITY;  // causal mask
        p[kt][e] = sv;
        rmax = fmaxf(rmax, sv);
      }
      rmax = group_reduce_max<16>(rmax);
      const float m_new = fmaxf(m_run[e], rmax);
      const float alpha = (m_run[e] == -INFINITY) ? 0.f : __expf(m_run[e] - m_new);
      float rsum = 0.f;
#pragma unroll
      for (int kt = 0; kt < 4; ++kt) {
        float pv = (p[kt][e] == -INFINITY) ? 0.f : __expf(p[kt][e] - m_new);
        p[kt][e] = pv;
        rsum += pv;
      }
      rsum = group_reduce_sum<16>(rsum);
      l_run[e] = l_run[e] * alpha + rsum;
      m_run[e] = m_new;
      // rescale O rows for this e
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) o_acc[nt][e] *= alpha;
    }

    // ---- P -> wave-private LDS tile (bf16, swizzled rows of 64) ----
#pragma unroll
    for (int kt = 0; kt < 4; ++kt) {
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int row = (lane >> 4) * 4 + e;
        const int col = (lane & 15) + kt * 16;
        const int phys = ((col >> 3) ^ (row & 7)) * 8 + (col & 7);
        p_lds[wid * 16 * KVBLK + row * KVBLK + phys] = f2bf(p[kt][e]);
      }
    }
    // wave-private tile: in-wave ds ordering is handled by the compiler's
    // lgkmcnt bookkeeping; no cross-wave barrier needed before the reads.

    // ---- PV: O[16,128] += P[16,64] @ V[64,128] (VT rows are k-contiguous) ----
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // kv 64 in 2 steps of 32
      short8 pfrag = *reinterpret_cast<const short8*>(
          p_lds + wid * 16 * KVBLK + ((lane & 15) * 8 + (((kk * 4 + kg) & 7) ^ ((lane & 15) & 7))) * 8);
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        short8 vfrag = read_swz(vt_lds, nt * 16 + (lane & 15), kk * 4 + kg, 8);
        o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, o_acc[nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_barrier();  // all reads of this tile done (every
    // ds_read's data was consumed by an MFMA, so lgkm already waited)
    cur ^= 1;
  }

  // ---- epilogue: O /= l, write bf16 ----
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    const int qrow = q0 + (lane >> 4) * 4 + e;
    const float inv_l = (l_run[e] > 0.f) ? 1.f / l_run[e] : 0.f;
    ushort* orow = Oh + (long long)qrow * D;
#pragma unroll
    for (int nt = 0; nt < 8; ++nt)
      orow[(lane & 15) + nt * 16] = f2bf(o_acc[nt][e] * inv_l);
  }
}
