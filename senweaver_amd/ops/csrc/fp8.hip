// fp8 (OCP e4m3) path for the 70B scorer (BASELINE config 5).
//
// gfx950 uses OCP e4m3fn (max 448), NOT MI300X's fnuz.  Weights are
// quantized once per-output-row, activations per-token-row on the fly;
// the GEMM runs v_mfma_f32_16x16x32_fp8_fp8 over a 128x128 tile with
// BK=128 (fp8 halves the staged bytes per K element, so the same 16 KiB
// LDS tile covers twice the K depth of the bf16 kernel) and the epilogue
// rescales by a_scale[m] * b_scale[n].
//
// Note on rates: non-MX-scaled fp8 MFMA runs at the bf16 rate on CDNA4
// (the 2x peak needs the MX block-scaled K=128 instructions — a follow-up);
// what this path buys today is halved weight/activation traffic and the
// fp8 numerics contract for ranking-stability testing.

#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;

#define FBK 128  // fp8 K-tile depth (bytes == elements)

// ---------------------------------------------------------------------------
// Row-wise quantization: bf16 [rows, K] -> fp8 bytes + f32 scale per row.
//   scale = amax/448; q = fp8(x/scale)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
quant_fp8_rowwise_kernel(const ushort* __restrict__ x, unsigned char* __restrict__ q,
                         float* __restrict__ scales, int K) {
  const long long base = (long long)blockIdx.x * K;
  float amax = 0.f;
  const int nvec = K / 8;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + base + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(bf2f(v.v[j])));
  }
  __shared__ float sm[4];
  float wmax = wave_reduce_max(amax);
  if ((threadIdx.x & 63) == 0) sm[threadIdx.x / 64] = wmax;
  __syncthreads();
  const float total = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
  const float scale = (total > 0.f) ? total / 448.f : 1.f;
  const float inv = 1.f / scale;
  if (threadIdx.x == 0) scales[blockIdx.x] = scale;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + base + i * 8);
    unsigned int lo = 0, hi = 0;
    lo = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v.v[0]) * inv, bf2f(v.v[1]) * inv, lo, false);
    lo = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v.v[2]) * inv, bf2f(v.v[3]) * inv, lo, true);
    hi = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v.v[4]) * inv, bf2f(v.v[5]) * inv, hi, false);
    hi = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f(v.v[6]) * inv, bf2f(v.v[7]) * inv, hi, true);
    uint2 out = {lo, hi};
    *reinterpret_cast<uint2*>(q + base + i * 8) = out;
  }
}

// ---------------------------------------------------------------------------
// fp8 GEMM: C[M,N] = (A_q[M,K] @ B_q[N,K]^T) * a_s[m] * b_s[n], bf16 out.
// 128x128 tile, 4 waves (2x2), glds double buffer, one barrier per K-tile.
// ---------------------------------------------------------------------------
__device__ __forceinline__ void stage_fp8_tile(
    const unsigned char* __restrict__ src, long long ldK, unsigned char* lds_tile,
    int tid) {
  // 128 rows x 8 chunks(16B=16 fp8) = 1024 chunks / 256 threads = 4 glds
  const int wave_chunk = tid & ~63;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int s = i * 256 + tid;
    const int row = s >> 3;
    const int c = (s & 7) ^ (row & 7);
    const unsigned char* g = src + (long long)row * ldK + c * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_tile + (long long)(i * 256 + wave_chunk) * 16),
        16, 0, 0);
  }
}

__device__ __forceinline__ long read_fp8_frag(const unsigned char* lds_tile, int row,
                                              int byte_off) {
  const int c = byte_off >> 4;
  const int phys = (c ^ (row & 7)) * 16 + (byte_off & 15);
  return *reinterpret_cast<const long*>(lds_tile + row * FBK + phys);
}

extern "C" __global__ void __launch_bounds__(256, 2)
gemm_bt_fp8_kernel(const unsigned char* __restrict__ A, const float* __restrict__ a_s,
                   const unsigned char* __restrict__ B, const float* __restrict__ b_s,
                   ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / 128) * (N / 128);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / 128;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1;
  const int wn = wid & 1;

  __shared__ __attribute__((aligned(16))) unsigned char lds[2][2][128 * FBK];

  const unsigned char* Atile = A + (long long)tile_m * 128 * K;
  const unsigned char* Btile = B + (long long)tile_n * 128 * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int m_base = wm * 64;
  const int n_base = wn * 64;
  const int frag_row = lane & 15;
  const int frag_kgrp = lane >> 4;

  const int ntiles = K / FBK;
  stage_fp8_tile(Atile, K, lds[0][0], tid);
  stage_fp8_tile(Btile, K, lds[0][1], tid);
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage_fp8_tile(Atile + (long long)(t + 1) * FBK, K, lds[buf ^ 1][0], tid);
      stage_fp8_tile(Btile + (long long)(t + 1) * FBK, K, lds[buf ^ 1][1], tid);
    }
    const unsigned char* Al = lds[buf][0];
    const unsigned char* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {  // FBK=128 in 4 MFMA steps of K=32
      long af[4], bf[4];
      const int boff = kk * 32 + frag_kgrp * 8;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = read_fp8_frag(Al, m_base + mi * 16 + frag_row, boff);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = read_fp8_frag(Bl, n_base + ni * 16 + frag_row, boff);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    buf ^= 1;
  }

  const long long c_row0 = (long long)tile_m * 128 + m_base + (lane >> 4) * 4;
  const long long c_col0 = (long long)tile_n * 128 + n_base + (lane & 15);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const long long row = c_row0 + mi * 16 + e;
      ushort* crow = C + row * N;
      const float as = a_s[row];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const long long col = c_col0 + ni * 16;
        crow[col] = f2bf(acc[mi][ni][e] * as * b_s[col]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// MX block-scaled fp8 path — the CDNA4 5-PF-class fp8 rate.
// v_mfma_scale_f32_32x32x64_f8f6f4 takes per-lane e8m0 scales covering each
// lane's 32-element K block (HW-fused dequant: D += (A*2^sa)@(B*2^sb)), so
// the epilogue needs NO rescale.  OCP MX quantization: per 32-element block,
// scale exponent = floor(log2(amax)) - 8 (e4m3 emax), elements saturate at
// +-448.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// bf16 [rows, K] -> fp8 q[rows, K] + e8m0 scales [rows, K/32]
extern "C" __global__ void __launch_bounds__(256)
quant_mxfp8_kernel(const ushort* __restrict__ x, unsigned char* __restrict__ q,
                   unsigned char* __restrict__ scales, int K) {
  const long long row = blockIdx.x;
  const int nblocks = K / 32;
  for (int blk = threadIdx.x; blk < nblocks; blk += blockDim.x) {
    const ushort* src = x + row * K + blk * 32;
    float v[32];
    float amax = 0.f;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 w = *reinterpret_cast<const bf16x8*>(src + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        v[c * 8 + j] = bf2f(w.v[j]);
        amax = fmaxf(amax, fabsf(v[c * 8 + j]));
      }
    }
    int e;
    if (amax > 0.f) {
      int ex;
      float m = frexpf(amax, &ex);  // amax = m * 2^ex, m in [0.5, 1)
      e = (ex - 1) - 8;             // floor(log2(amax)) - emax(e4m3)
      // non-saturating: if amax/2^e = m*512 > 448, step one finer
      if (m > 0.875f) e += 1;
    } else {
      e = -127;
    }
    e = max(-127, min(127, e));
    scales[row * nblocks + blk] = (unsigned char)(e + 127);
    const float inv = exp2f((float)(-e));
    unsigned char* dst = q + row * K + blk * 32;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      unsigned w = 0;
      w = __builtin_amdgcn_cvt_pk_fp8_f32(v[c * 4 + 0] * inv, v[c * 4 + 1] * inv, w, false);
      w = __builtin_amdgcn_cvt_pk_fp8_f32(v[c * 4 + 2] * inv, v[c * 4 + 3] * inv, w, true);
      *reinterpret_cast<unsigned*>(dst + c * 4) = w;
    }
  }
}

// C[M,N] = dequant(A_q, A_s) @ dequant(B_q, B_s)^T, bf16 out.
// 128x128 tile, BK=128 fp8 bytes, 4 waves (2x2, 64x64 per wave = 2x2 frags
// of 32x32), glds double buffer + one __syncthreads per K-tile.
//
// Fragment layout (probed empirically, benchmarks/probe_mx.py): per 64-k
// instruction, lane half lhi holds k in [lhi*16, lhi*16+16) u
// [32+lhi*16, 32+lhi*16+16) — 16 bytes from EACH of the two 32-element
// scale blocks — while the per-lane scale register of half lhi supplies
// the e8m0 scale for block lhi.  So the two 16-B chunks a lane reads are
// (kk*4 + lhi) and (kk*4 + lhi + 2), and the scale byte is block
// (t*4 + kk*2 + lhi) of that fragment's row.
__device__ __forceinline__ i32x8 read_mx_frag(const unsigned char* lds_tile,
                                              int row, int c0) {
  // two 16-B chunks (c0, c0+2) — one from each scale block — XOR-swizzled
  i32x8 out;
  const int p0 = (c0 ^ (row & 7)) * 16;
  const int p1 = ((c0 + 2) ^ (row & 7)) * 16;
  *reinterpret_cast<int4*>(&out) = *reinterpret_cast<const int4*>(lds_tile + row * FBK + p0);
  *(reinterpret_cast<int4*>(&out) + 1) = *reinterpret_cast<const int4*>(lds_tile + row * FBK + p1);
  return out;
}

extern "C" __global__ void __launch_bounds__(256, 2)
gemm_bt_mxfp8_kernel(const unsigned char* __restrict__ A, const unsigned char* __restrict__ As,
                     const unsigned char* __restrict__ B, const unsigned char* __restrict__ Bs,
                     ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / 128) * (N / 128);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / 128;
  const int tile_m = wgid / tiles_n;
  const int tile_n = wgid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1;
  const int wn = wid & 1;
  const int l31 = lane & 31;
  const int lhi = lane >> 5;

  __shared__ __attribute__((aligned(16))) unsigned char lds[2][2][128 * FBK];

  const unsigned char* Atile = A + (long long)tile_m * 128 * K;
  const unsigned char* Btile = B + (long long)tile_n * 128 * K;
  const int sld = K / 32;  // scale row stride
  const unsigned char* Astile = As + (long long)tile_m * 128 * sld;
  const unsigned char* Bstile = Bs + (long long)tile_n * 128 * sld;

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;

  const int m_base = wm * 64;
  const int n_base = wn * 64;

  const int ntiles = K / FBK;
  stage_fp8_tile(Atile, K, lds[0][0], tid);
  stage_fp8_tile(Btile, K, lds[0][1], tid);

  // scale rows this lane's fragments use (dword = the 4 e8m0 bytes of one
  // 128-deep K-tile); prefetched one tile ahead so the byte extracts in the
  // MFMA loop never wait on global memory
  const unsigned char* As_row[2] = {
      Astile + (long long)(m_base + 0 * 32 + l31) * sld,
      Astile + (long long)(m_base + 1 * 32 + l31) * sld};
  const unsigned char* Bs_row[2] = {
      Bstile + (long long)(n_base + 0 * 32 + l31) * sld,
      Bstile + (long long)(n_base + 1 * 32 + l31) * sld};
  unsigned sa_dw[2], sb_dw[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    sa_dw[i] = *reinterpret_cast<const unsigned*>(As_row[i]);
    sb_dw[i] = *reinterpret_cast<const unsigned*>(Bs_row[i]);
  }
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    unsigned na[2], nb[2];
    if (t + 1 < ntiles) {
      stage_fp8_tile(Atile + (long long)(t + 1) * FBK, K, lds[buf ^ 1][0], tid);
      stage_fp8_tile(Btile + (long long)(t + 1) * FBK, K, lds[buf ^ 1][1], tid);
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        na[i] = *reinterpret_cast<const unsigned*>(As_row[i] + (t + 1) * 4);
        nb[i] = *reinterpret_cast<const unsigned*>(Bs_row[i] + (t + 1) * 4);
      }
    }
    const unsigned char* Al = lds[buf][0];
    const unsigned char* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // FBK=128 in 2 MFMA steps of K=64
      // lane half lhi supplies the scale of block (kk*2 + lhi) of this tile
      const int sh = 8 * (kk * 2 + lhi);
      i32x8 af[2], bf[2];
      int sa[2], sb[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        af[mi] = read_mx_frag(Al, m_base + mi * 32 + l31, kk * 4 + lhi);
        sa[mi] = (sa_dw[mi] >> sh) & 0xff;
      }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        bf[ni] = read_mx_frag(Bl, n_base + ni * 32 + l31, kk * 4 + lhi);
        sb[ni] = (sb_dw[ni] >> sh) & 0xff;
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0, sa[mi], 0, sb[ni]);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    buf ^= 1;
#pragma unroll
    for (int i = 0; i < 2; ++i) { sa_dw[i] = na[i]; sb_dw[i] = nb[i]; }
  }

  // C/D 32x32 map: col = lane&31, row = (r&3)+8*(r>>2)+4*(lane>>5)
  const long long c_col0 = (long long)tile_n * 128 + n_base + l31;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long long row = (long long)tile_m * 128 + m_base + mi * 32
                            + (r & 3) + 8 * (r >> 2) + 4 * lhi;
      ushort* crow = C + row * N;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        crow[c_col0 + ni * 32] = f2bf(acc[mi][ni][r]);
    }
}

// ---------------------------------------------------------------------------
// 256x256-tile MX fp8 GEMM: 8 waves (2 m x 4 n), wave tile 128x64 as 4x2
// frags of 32x32.  Raises the MFMA:ds_read ratio from 8:4 to 8:6-per-
// double-the-work (6 frag reads feed 8 scaled MFMAs vs 4 feeding 4) and
// halves barriers per flop; 1 block/CU (acc alone is 128 VGPRs).
// ---------------------------------------------------------------------------
__device__ __forceinline__ void stage_fp8_tile_256(
    const unsigned char* __restrict__ src, long long ldK, unsigned char* lds_tile,
    int tid) {
  // 256 rows x 8 chunks(16B) = 2048 chunks / 512 threads = 4 glds each
  const int wave_chunk = tid & ~63;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int s = i * 512 + tid;
    const int row = s >> 3;
    const int c = (s & 7) ^ (row & 7);
    const unsigned char* g = src + (long long)row * ldK + c * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds_tile + (long long)(i * 512 + wave_chunk) * 16),
        16, 0, 0);
  }
}

extern "C" __global__ void __launch_bounds__(512, 1)
gemm_bt_mxfp8_256_kernel(const unsigned char* __restrict__ A, const unsigned char* __restrict__ As,
                         const unsigned char* __restrict__ B, const unsigned char* __restrict__ Bs,
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
  const int wm = wid >> 2;      // 0..1 -> 128-row half
  const int wn = wid & 3;       // 0..3 -> 64-col quarter
  const int l31 = lane & 31;
  const int lhi = lane >> 5;

  __shared__ __attribute__((aligned(16))) unsigned char lds[2][2][256 * FBK];

  const unsigned char* Atile = A + (long long)tile_m * 256 * K;
  const unsigned char* Btile = B + (long long)tile_n * 256 * K;
  const int sld = K / 32;

  f32x16 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;

  const int m_base = wm * 128;
  const int n_base = wn * 64;

  const unsigned char* As_row[4];
  const unsigned char* Bs_row[2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    As_row[i] = As + ((long long)tile_m * 256 + m_base + i * 32 + l31) * sld;
#pragma unroll
  for (int i = 0; i < 2; ++i)
    Bs_row[i] = Bs + ((long long)tile_n * 256 + n_base + i * 32 + l31) * sld;
  unsigned sa_dw[4], sb_dw[2];
#pragma unroll
  for (int i = 0; i < 4; ++i) sa_dw[i] = *reinterpret_cast<const unsigned*>(As_row[i]);
#pragma unroll
  for (int i = 0; i < 2; ++i) sb_dw[i] = *reinterpret_cast<const unsigned*>(Bs_row[i]);

  const int ntiles = K / FBK;
  stage_fp8_tile_256(Atile, K, lds[0][0], tid);
  stage_fp8_tile_256(Btile, K, lds[0][1], tid);
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < ntiles; ++t) {
    unsigned na[4], nb[2];
    if (t + 1 < ntiles) {
      stage_fp8_tile_256(Atile + (long long)(t + 1) * FBK, K, lds[buf ^ 1][0], tid);
      stage_fp8_tile_256(Btile + (long long)(t + 1) * FBK, K, lds[buf ^ 1][1], tid);
#pragma unroll
      for (int i = 0; i < 4; ++i) na[i] = *reinterpret_cast<const unsigned*>(As_row[i] + (t + 1) * 4);
#pragma unroll
      for (int i = 0; i < 2; ++i) nb[i] = *reinterpret_cast<const unsigned*>(Bs_row[i] + (t + 1) * 4);
    }
    const unsigned char* Al = lds[buf][0];
    const unsigned char* Bl = lds[buf][1];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int sh = 8 * (kk * 2 + lhi);
      i32x8 af[4], bf[2];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = read_mx_frag(Al, m_base + mi * 32 + l31, kk * 4 + lhi);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        bf[ni] = read_mx_frag(Bl, n_base + ni * 32 + l31, kk * 4 + lhi);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0,
              (int)((sa_dw[mi] >> sh) & 0xff), 0, (int)((sb_dw[ni] >> sh) & 0xff));
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    buf ^= 1;
#pragma unroll
    for (int i = 0; i < 4; ++i) sa_dw[i] = na[i];
#pragma unroll
    for (int i = 0; i < 2; ++i) sb_dw[i] = nb[i];
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
// MX fp8 16-wave single-barrier pipeline — the v14 bf16 structure
// (gemm_pipe.hip) at the 2x block-scaled MFMA rate.
//
// 1024 threads / 16 waves (4 per SIMD), wave tile 64x64 as 2x2 fragments of
// v_mfma_scale_f32_32x32x64_f8f6f4; FBK=128 K-tile = one 32 KiB stage unit
// per operand (256 rows x 128 fp8 = full 128-B rows); A double-buffered +
// B ring-of-3 across the whole 160 KiB LDS; ONE barrier per K-tile with the
// counted vmcnt placed BEFORE it (wait-then-rendezvous).
//
// LDS swizzle: chunk c of row r at slot (c + (r>>1)) & 7 — conflict-free
// for the 32-row fragment read pattern (row = base + lane&31, chunks
// kk*4+lhi and +2): within each ds_read_b128 16-lane group the same-parity
// rows hit 8 distinct 16-B slots (derived against the (addr/4)%64 banking).
//
// e8m0 scales ([row][K/32] bytes; 1 dword covers one K-tile's 4 blocks) are
// prefetched one tile ahead with INLINE-ASM global_load_dword: hipcc's wait
// bookkeeping cannot see them, so they never trigger the
// vmcnt(0)-at-ordinary-load-use drain that would serialize the glds
// pipeline; the tile-boundary s_waitcnt vmcnt(2) (which names the scale
// registers as operands) is what guarantees they have landed.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(1024, 4)
gemm_bt_mxfp8_pipe_kernel(const unsigned char* __restrict__ A, const unsigned char* __restrict__ As,
                          const unsigned char* __restrict__ B, const unsigned char* __restrict__ Bs,
                          ushort* __restrict__ C, int M, int N, int K) {
  const int nwg = (M / 256) * (N / 256);
  int wgid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wgid % 8, pos = wgid / 8;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / 256;
  const int GM = 8;
  const int tiles_m = M / 256;
  const int grp = wgid / (GM * tiles_n);
  const int rem = wgid % (GM * tiles_n);
  const int g0 = grp * GM;
  const int gh = (tiles_m - g0 < GM) ? (tiles_m - g0) : GM;
  const int tile_m = g0 + rem % gh;
  const int tile_n = rem / gh;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;       // 0..15
  const int wm = wid >> 2;        // 0..3 -> 64-row band of A
  const int wn = wid & 3;         // 0..3 -> 64-row band of B
  const int l31 = lane & 31;
  const int lhi = lane >> 5;

  // 5 x 32 KiB: A in slots 0,1 (dbuf); B ring in 2,3,4
  __shared__ __attribute__((aligned(16))) unsigned char lds[5][256 * FBK];

  const unsigned char* Atile = A + (long long)tile_m * 256 * K;
  const unsigned char* Btile = B + (long long)tile_n * 256 * K;
  const int sld = K / 32;

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;

  // fragment-read swizzle constant: (c + (row>>1)) & 7 with row = base+l31,
  // base % 32 == 0 -> per-lane constant part is (l31>>1) & 7
  const int swp = (l31 >> 1) & 7;
  const int a_row0 = wm * 64;   // + mi*32 + l31
  const int b_row0 = wn * 64;

  // staging source coords (i = 0..3): slot s = i*512 + tid, row = s>>3,
  // dest chunk sc = s&7; source chunk = (sc - (row>>1)) & 7.  Kept as ONE
  // 32-bit byte offset per i (row*K + chunk*16): 4 VGPRs total instead of
  // four 64-bit pointer pairs, which spilled into the K-loop.
  unsigned st_off[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int s = i * 1024 + tid;
    const int r = s >> 3;
    st_off[i] = (unsigned)r * (unsigned)K + ((((s & 7) - (r >> 1)) & 7) * 16);
  }
  const int wave_chunk = tid & ~63;

  const int ntiles = K / FBK;

#define MXISSUE(TGT, OP, SLOT)                                               \
  do {                                                                       \
    if ((TGT) < ntiles) {                                                    \
      const int k0_ = (TGT) * FBK;                                           \
      unsigned char* dst_ = &lds[(SLOT)][0];                                 \
      const unsigned char* opk_ = (OP) + k0_;                                \
      _Pragma("unroll") for (int i = 0; i < 2; ++i) {                        \
        const unsigned char* g = opk_ + st_off[i];                           \
        __builtin_amdgcn_global_load_lds(                                    \
            (const __attribute__((address_space(1))) unsigned int*)g,        \
            (__attribute__((address_space(3))) unsigned int*)(dst_ +         \
                (long long)(i * 1024 + wave_chunk) * 16),                    \
            16, 0, 0);                                                       \
      }                                                                      \
    }                                                                        \
  } while (0)

  // Scale access via SRSRC buffer loads (T8/T20): the descriptor lives in
  // SGPRs (built from readfirstlane'd kernarg pointers, provably uniform),
  // each lane carries only a 32-bit voffset -> 4 VGPRs instead of 4
  // pointer pairs.  sld bytes per row; one dword per K-tile (FBK=128).
  const unsigned as_bytes = (unsigned)M * (unsigned)sld;
  const unsigned bs_bytes = (unsigned)N * (unsigned)sld;
  const auto rsrc_a = __builtin_amdgcn_make_buffer_rsrc(
      (void*)As, 0, as_bytes, 0x00020000);
  const auto rsrc_b = __builtin_amdgcn_make_buffer_rsrc(
      (void*)Bs, 0, bs_bytes, 0x00020000);
  unsigned voa[2], vob[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    voa[i] = (unsigned)((tile_m * 256 + a_row0 + i * 32 + l31) * sld);
    vob[i] = (unsigned)((tile_n * 256 + b_row0 + i * 32 + l31) * sld);
  }
  // tile-0 scales with plain buffer loads (before any glds is outstanding)
  unsigned sa_cur[2], sb_cur[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    sa_cur[i] = __builtin_amdgcn_raw_buffer_load_b32(rsrc_a, voa[i], 0, 0);
    sb_cur[i] = __builtin_amdgcn_raw_buffer_load_b32(rsrc_b, vob[i], 0, 0);
  }

  MXISSUE(0, Btile, 2);
  MXISSUE(0, Atile, 0);
  MXISSUE(1, Btile, 3);
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  __builtin_amdgcn_s_barrier();

#define MXFRAG(DST, SLOT, ROW, C0)                                           \
  do {                                                                       \
    const unsigned char* base_ = &lds[(SLOT)][0] + (long long)(ROW) * FBK;   \
    const int p0_ = ((((C0) & 7) + swp) & 7) * 16;                           \
    const int p1_ = (((((C0) + 2) & 7) + swp) & 7) * 16;                     \
    *reinterpret_cast<int4*>(&DST) =                                         \
        *reinterpret_cast<const int4*>(base_ + p0_);                         \
    *(reinterpret_cast<int4*>(&DST) + 1) =                                   \
        *reinterpret_cast<const int4*>(base_ + p1_);                         \
  } while (0)

  for (int t = 0; t < ntiles; ++t) {
    const int aslot = t & 1;
    const int bslot = 2 + t % 3;
    const int bslot2 = 2 + (t + 2) % 3;
    // prefetch next tile's scale dwords via asm loads (invisible to hipcc's
    // waitcnt bookkeeping -> no pipeline-draining use-waits); drained by the
    // boundary vmcnt(2) below, which names them as operands.
    unsigned sa_n[2], sb_n[2];
    if (t + 1 < ntiles) {
      const unsigned so = (unsigned)((t + 1) * 4);
      asm volatile("buffer_load_dword %0, %4, %8, %10 offen\n\t"
                   "buffer_load_dword %1, %5, %8, %10 offen\n\t"
                   "buffer_load_dword %2, %6, %9, %10 offen\n\t"
                   "buffer_load_dword %3, %7, %9, %10 offen"
                   : "=&v"(sa_n[0]), "=&v"(sa_n[1]), "=&v"(sb_n[0]), "=&v"(sb_n[1])
                   : "v"(voa[0]), "v"(voa[1]), "v"(vob[0]), "v"(vob[1]),
                     "s"(rsrc_a), "s"(rsrc_b), "s"(so)
                   : "memory");
    } else {
      sa_n[0] = sa_n[1] = sb_n[0] = sb_n[1] = 0;
    }
    i32x8 af, bf[2];
    // kk = 0 (K 0..63 of the tile)
    MXFRAG(bf[0], bslot, b_row0 + 0 * 32 + l31, 0 * 4 + lhi);
    MXFRAG(bf[1], bslot, b_row0 + 1 * 32 + l31, 0 * 4 + lhi);
    MXISSUE(t + 1, Atile, aslot ^ 1);
    {
      const int sh = 8 * (0 * 2 + lhi);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        MXFRAG(af, aslot, a_row0 + mi * 32 + l31, 0 * 4 + lhi);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af, bf[ni], acc[mi][ni], 0, 0, 0,
              (int)((sa_cur[mi] >> sh) & 0xff), 0,
              (int)((sb_cur[ni] >> sh) & 0xff));
      }
    }
    // kk = 1 (K 64..127)
    MXFRAG(bf[0], bslot, b_row0 + 0 * 32 + l31, 1 * 4 + lhi);
    MXFRAG(bf[1], bslot, b_row0 + 1 * 32 + l31, 1 * 4 + lhi);
    MXISSUE(t + 2, Btile, bslot2);
    {
      const int sh = 8 * (1 * 2 + lhi);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        MXFRAG(af, aslot, a_row0 + mi * 32 + l31, 1 * 4 + lhi);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af, bf[ni], acc[mi][ni], 0, 0, 0,
              (int)((sa_cur[mi] >> sh) & 0xff), 0,
              (int)((sb_cur[ni] >> sh) & 0xff));
      }
    }
    // boundary: wait (drains A(t+1), B(t+1) glds AND the scale loads; only
    // B(t+2)'s 2 glds stay in flight), then rendezvous
    if (t >= ntiles - 2)
      asm volatile("s_waitcnt vmcnt(0)"
                   : "+v"(sa_n[0]), "+v"(sa_n[1]), "+v"(sb_n[0]), "+v"(sb_n[1])
                   :: "memory");
    else
      asm volatile("s_waitcnt vmcnt(2)"
                   : "+v"(sa_n[0]), "+v"(sa_n[1]), "+v"(sb_n[0]), "+v"(sb_n[1])
                   :: "memory");
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int i = 0; i < 2; ++i) { sa_cur[i] = sa_n[i]; sb_cur[i] = sb_n[i]; }
  }
#undef MXFRAG
#undef MXISSUE

  // 32x32 C map: col = l31 (+ni*32), row = (r&3) + 8*(r>>2) + 4*lhi
  const long long c_col0 = (long long)tile_n * 256 + b_row0 + l31;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long long row = (long long)tile_m * 256 + a_row0 + mi * 32
                            + (r & 3) + 8 * (r >> 2) + 4 * lhi;
      ushort* crow = C + row * N;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        crow[c_col0 + ni * 32] = f2bf(acc[mi][ni][r]);
    }
}
