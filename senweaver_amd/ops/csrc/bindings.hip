// PyTorch bindings for the senweaver_amd gfx950 kernel library.
// HIP-native throughout (c10::hip stream accessors) — no CUDA shims.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "common.h"

// ---- kernel prototypes (defined in the sibling .hip translation units) ----
extern "C" __global__ void rmsnorm_fwd_kernel(const ushort*, const ushort*, ushort*, ushort*, int, float);
extern "C" __global__ void rope_fwd_kernel(ushort*, ushort*, const float*, const int*, int, int, int);
extern "C" __global__ void rope_scatter_kernel(const ushort*, long long, ushort*, ushort*, const float*, const int*, int, int, int, int);
extern "C" __global__ void transpose_v_kernel(const ushort*, long long, long long, ushort*, int, int, int, int);
extern "C" __global__ void rope_kv_append_kernel(const ushort*, long long, ushort*, ushort*, ushort*, const float*, const int*, const int*, int, int, int);
extern "C" __global__ void swiglu_fwd_kernel(const ushort*, ushort*, long long, int);
extern "C" __global__ void add_bf16_kernel(const ushort*, const ushort*, ushort*, long long);
extern "C" __global__ void argmax_rows_kernel(const ushort*, int*, int);
extern "C" __global__ void target_logprob_kernel(const ushort*, const int*, float*, int);
extern "C" __global__ void gemm_bt_bf16_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_256_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_256x32_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_256sg_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v1_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v2_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v3_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v4_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v5_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v6_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v7_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v8_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v9_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v11_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v12_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v13_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v14_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v16_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_8ph_v17_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_asm_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_asm2_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_asm3_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_asm4_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_asm5_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_asm6_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_bf16_asm7_kernel(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void grouped_gemm_bt_bf16_kernel(const ushort*, const ushort*, ushort*, const int*, const int*, const int*, int, int, int);
extern "C" __global__ void moe_combine_kernel(const ushort*, const int*, const float*, ushort*, int, int);
extern "C" __global__ void quant_fp8_rowwise_kernel(const ushort*, unsigned char*, float*, int);
extern "C" __global__ void gemm_bt_fp8_kernel(const unsigned char*, const float*, const unsigned char*, const float*, ushort*, int, int, int);
extern "C" __global__ void quant_mxfp8_kernel(const ushort*, unsigned char*, unsigned char*, int);
extern "C" __global__ void gemm_bt_mxfp8_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_mxfp8_256_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, ushort*, int, int, int);
extern "C" __global__ void gemm_bt_mxfp8_pipe_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_bf16_v2_m1(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_bf16_v3_m1(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_fp8w_m1(const ushort*, const unsigned char*, const float*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_fp8w_m2(const ushort*, const unsigned char*, const float*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_fp8w_m4(const ushort*, const unsigned char*, const float*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_fp8w_m8(const ushort*, const unsigned char*, const float*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_mxfp8w_m1(const ushort*, const unsigned char*, const unsigned char*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_mxfp8w_m2(const ushort*, const unsigned char*, const unsigned char*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_mxfp8w_m4(const ushort*, const unsigned char*, const unsigned char*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_mxfp8w_m8(const ushort*, const unsigned char*, const unsigned char*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_bf16_v3w_m1(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void swiglu_gemv_bt_bf16_m1(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_bf16_v2_m2(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_bf16_v2_m4(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_bf16_v2_m8(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemv_bt_bf16_v2_m16(const ushort*, const ushort*, ushort*, int, int, int);
extern "C" __global__ void gemv_norm_bt_bf16_m1(const ushort*, const ushort*, const ushort*, const ushort*, ushort*, ushort*, int, int, int, float);
extern "C" __global__ void gemv_norm_bt_bf16_m2(const ushort*, const ushort*, const ushort*, const ushort*, ushort*, ushort*, int, int, int, float);
extern "C" __global__ void gemv_norm_bt_bf16_m4(const ushort*, const ushort*, const ushort*, const ushort*, ushort*, ushort*, int, int, int, float);
extern "C" __global__ void gemv_norm_bt_bf16_m8(const ushort*, const ushort*, const ushort*, const ushort*, ushort*, ushort*, int, int, int, float);
extern "C" __global__ void gemv_norm_bt_bf16_m16(const ushort*, const ushort*, const ushort*, const ushort*, ushort*, ushort*, int, int, int, float);
extern "C" __global__ void attn_fwd_bf16_kernel(const ushort*, const ushort*, const ushort*, ushort*, int, int, int, int, float);
extern "C" __global__ void attn_fwd_v2_kernel(const ushort*, const ushort*, const ushort*, ushort*, int, int, int, int, float);
extern "C" __global__ void attn_fwd_v3_kernel(const ushort*, const ushort*, const ushort*, ushort*, int, int, int, int, float);
extern "C" __global__ void paged_decode_attn_partial_kernel(const ushort*, const ushort*, const ushort*, float*,
                                                            const int*, const int*, int, int, int, float);
extern "C" __global__ void paged_decode_attn_merge_kernel(const float*, ushort*, int);

namespace {

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

inline const ushort* bf16_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const ushort*>(t.data_ptr());
}
inline ushort* bf16_mut(torch::Tensor& t) {
  return reinterpret_cast<ushort*>(t.data_ptr());
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

}  // namespace

// ---------------- RMSNorm ----------------
torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor w, double eps) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden % 8 != 0");
  const long long rows = x.numel() / hidden;
  auto y = torch::empty_like(x);
  rmsnorm_fwd_kernel<<<dim3((unsigned)rows), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(x), bf16_ptr(w), bf16_mut(y), nullptr, hidden, (float)eps);
  HIP_CHECK_KERNEL();
  return y;
}

// y = rmsnorm(x + residual); residual <- x + residual (in place)
torch::Tensor fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                                torch::Tensor w, double eps) {
  check_bf16(x, "x");
  check_bf16(residual, "residual");
  check_bf16(w, "w");
  const int hidden = x.size(-1);
  const long long rows = x.numel() / hidden;
  auto y = torch::empty_like(x);
  rmsnorm_fwd_kernel<<<dim3((unsigned)rows), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(x), bf16_ptr(w), bf16_mut(y), bf16_mut(residual), hidden, (float)eps);
  HIP_CHECK_KERNEL();
  return y;
}

// ---------------- RoPE (in place over q and k) ----------------
void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos_sin,
                  torch::Tensor positions) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32 && cos_sin.is_contiguous());
  TORCH_CHECK(positions.scalar_type() == torch::kInt32 && positions.is_contiguous());
  const int T = q.size(0);
  const int Hq = q.size(1), Hk = k.size(1), D = q.size(2);
  TORCH_CHECK(k.size(0) == T && k.size(2) == D);
  TORCH_CHECK(cos_sin.size(1) == D);
  rope_fwd_kernel<<<dim3(T, Hq + Hk), dim3(64), 0, cur_stream()>>>(
      bf16_mut(q), bf16_mut(k), cos_sin.data_ptr<float>(),
      positions.data_ptr<int>(), Hq, Hk, D);
  HIP_CHECK_KERNEL();
}

// qkv: [T, Hq*D + 2*Hk*D] fused projection -> rope -> q [B,Hq,S,D], k [B,Hk,S,D]
std::vector<torch::Tensor> rope_scatter_qkv(torch::Tensor qkv, torch::Tensor cos_sin,
                                            torch::Tensor positions,
                                            int64_t Hq, int64_t Hk, int64_t D,
                                            int64_t B, int64_t S) {
  check_bf16(qkv, "qkv");
  const long long T = qkv.size(0);
  const long long ld = qkv.size(1);
  TORCH_CHECK(T == B * S && cos_sin.size(1) == D && ld >= (Hq + 2 * Hk) * D);
  auto qo = torch::empty({B, Hq, S, D}, qkv.options());
  auto ko = torch::empty({B, Hk, S, D}, qkv.options());
  TORCH_CHECK((D / 2) % 8 == 0, "rope_scatter needs D/2 % 8 == 0");
  const int heads_per_blk = (int)(256 / (D / 16));
  rope_scatter_kernel<<<dim3((unsigned)T, (unsigned)((Hq + Hk + heads_per_blk - 1) / heads_per_blk)),
                        dim3(256), 0, cur_stream()>>>(
      bf16_ptr(qkv), ld, bf16_mut(qo), bf16_mut(ko),
      cos_sin.data_ptr<float>(), positions.data_ptr<int>(), (int)Hq, (int)Hk,
      (int)D, (int)S);
  HIP_CHECK_KERNEL();
  return {qo, ko};
}

torch::Tensor vt_from_qkv(torch::Tensor qkv, int64_t Hq, int64_t Hk, int64_t D,
                          int64_t B, int64_t S) {
  check_bf16(qkv, "qkv");
  const long long ld = qkv.size(1);
  TORCH_CHECK(qkv.size(0) == B * S && ld >= (Hq + 2 * Hk) * D);
  auto vt = torch::empty({B, Hk, D, S}, qkv.options());
  const long long v_off = (Hq + Hk) * D;
  transpose_v_kernel<<<dim3((unsigned)((S + 31) / 32), (unsigned)((D + 63) / 64),
                           (unsigned)(B * Hk)),
                       dim3(256), 0, cur_stream()>>>(
      bf16_ptr(qkv), ld, v_off, bf16_mut(vt), (int)B, (int)S, (int)Hk, (int)D);
  HIP_CHECK_KERNEL();
  return vt;
}

torch::Tensor rope_kv_append(torch::Tensor qkv, torch::Tensor cos_sin,
                             torch::Tensor positions, torch::Tensor slot,
                             torch::Tensor kcache, torch::Tensor vcache,
                             int64_t Hq, int64_t Hk, int64_t D) {
  check_bf16(qkv, "qkv");
  const int B = qkv.size(0);
  const long long ld = qkv.size(1);
  TORCH_CHECK(ld >= (Hq + 2 * Hk) * D && (D / 2) % 8 == 0);
  TORCH_CHECK(positions.scalar_type() == torch::kInt32 &&
              slot.scalar_type() == torch::kInt32);
  auto q = torch::empty({(long)B, Hq, D}, qkv.options());
  const int heads_per_blk = (int)(256 / (D / 16));
  const long long H_total = Hq + 2 * Hk;
  rope_kv_append_kernel<<<dim3((unsigned)B, (unsigned)((H_total + heads_per_blk - 1) / heads_per_blk)),
                          dim3(256), 0, cur_stream()>>>(
      bf16_ptr(qkv), ld, bf16_mut(q),
      reinterpret_cast<ushort*>(kcache.data_ptr()),
      reinterpret_cast<ushort*>(vcache.data_ptr()),
      cos_sin.data_ptr<float>(), positions.data_ptr<int>(),
      slot.data_ptr<int>(), (int)Hq, (int)Hk, (int)D);
  HIP_CHECK_KERNEL();
  return q;
}

// ---------------- SwiGLU ----------------
torch::Tensor swiglu(torch::Tensor gateup) {
  check_bf16(gateup, "gateup");
  const int inter2 = gateup.size(-1);
  TORCH_CHECK(inter2 % 16 == 0);
  const int inter = inter2 / 2;
  const long long rows = gateup.numel() / inter2;
  auto sizes = gateup.sizes().vec();
  sizes.back() = inter;
  auto y = torch::empty(sizes, gateup.options());
  const int nvec_blocks = (inter / 8 + 255) / 256;
  swiglu_fwd_kernel<<<dim3(nvec_blocks, (unsigned)rows), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(gateup), bf16_mut(y), rows, inter);
  HIP_CHECK_KERNEL();
  return y;
}

torch::Tensor add_bf16(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  TORCH_CHECK(a.numel() == b.numel() && a.numel() % 8 == 0);
  auto y = torch::empty_like(a);
  const long long n8 = a.numel() / 8;
  const int blocks = (int)std::min<long long>((n8 + 255) / 256, 2048);
  add_bf16_kernel<<<dim3(blocks), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(a), bf16_ptr(b), bf16_mut(y), n8);
  HIP_CHECK_KERNEL();
  return y;
}

// ---------------- GEMM: C[M,N] = A[M,K] @ B[N,K]^T ----------------
torch::Tensor gemm_bt(torch::Tensor A, torch::Tensor B) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  auto C = torch::empty({M, N}, A.options());
  if (M <= 16) {
    TORCH_CHECK(N % 4 == 0 && K % 512 == 0, "gemv path needs N%4==0, K%512==0");
    // wave-per-row coalesced GEMV; A must be padded to the MM bucket rows
    int MM = M <= 1 ? 1 : M <= 2 ? 2 : M <= 4 ? 4 : M <= 8 ? 8 : 16;
    torch::Tensor Ap = A;
    if (A.size(0) < MM) {
      Ap = torch::zeros({MM, (long)K}, A.options());
      Ap.narrow(0, 0, M).copy_(A);
    }
    const dim3 grid((N + 3) / 4);
    auto launch = [&](auto kern) {
      kern<<<grid, dim3(256), 0, cur_stream()>>>(bf16_ptr(Ap), bf16_ptr(B),
                                                 bf16_mut(C), M, N, K);
    };
    // M=1 wide-K rows: the loader/consumer LDS-DMA streaming engine
    // measures +25% over v2 at K=14336 (5.9 vs 4.7 TB/s,
    // profiles/r02_gemv_engine.txt); at K<=4096 its per-block prologue
    // (x stage + 2-step ring fill) can't amortize over 4 steps and v2 wins
    if (MM == 1 && K % 1024 == 0 && K > 4096 && K <= 14336 && N % 8 == 0) {
      gemv_bt_bf16_v3w_m1<<<dim3(N / 8), dim3(256), 0, cur_stream()>>>(
          bf16_ptr(Ap), bf16_ptr(B), bf16_mut(C), M, N, K);
      HIP_CHECK_KERNEL();
      return C;
    }
    switch (MM) {
      case 1: launch(gemv_bt_bf16_v2_m1); break;
      case 2: launch(gemv_bt_bf16_v2_m2); break;
      case 4: launch(gemv_bt_bf16_v2_m4); break;
      case 8: launch(gemv_bt_bf16_v2_m8); break;
      default: launch(gemv_bt_bf16_v2_m16); break;
    }
    HIP_CHECK_KERNEL();
    return C;
  }
  TORCH_CHECK(K % 64 == 0, "gemm_bt needs K % 64 == 0; got K=", K);
  // 256-tile kernels run 1 block/CU (128 KiB LDS): they need >=~160 blocks
  // to fill 256 CUs; below that the 2-block/CU 128-tile kernel wins.
  if (M % 256 == 0 && N % 256 == 0 && (M / 256) * (N / 256) >= 160) {
    const int nwg = (M / 256) * (N / 256);
    if (K % 128 == 0)
      // in-repo best per shape (profiles/r02_gemm_pipeline.txt +
      // r02_gemm_asm_glds.txt): asm7 (raw glds staged INSIDE the MFMA
      // stream) wins big-M and wide-N shapes (1.41/1.23 PF at 8k/gateup);
      // asm4 (between-statement staging) keeps 4096-class; the 16-wave
      // plain-HIP pipeline keeps the remaining small-M shapes
      if (M >= 8192 || (long long)N * 2 >= (long long)M * 7)
        gemm_bt_bf16_asm7_kernel<<<dim3(nwg), dim3(512), 0, cur_stream()>>>(
            bf16_ptr(A), bf16_ptr(B), bf16_mut(C), M, N, K);
      else if (M >= 4096)
        gemm_bt_bf16_asm4_kernel<<<dim3(nwg), dim3(512), 0, cur_stream()>>>(
            bf16_ptr(A), bf16_ptr(B), bf16_mut(C), M, N, K);
      else
        gemm_bt_bf16_8ph_v14_kernel<<<dim3(nwg), dim3(1024), 0, cur_stream()>>>(
            bf16_ptr(A), bf16_ptr(B), bf16_mut(C), M, N, K);
    else
      gemm_bt_bf16_256_kernel<<<dim3(nwg), dim3(512), 0, cur_stream()>>>(
          bf16_ptr(A), bf16_ptr(B), bf16_mut(C), M, N, K);
  } else {
    TORCH_CHECK(M % 128 == 0 && N % 128 == 0,
                "gemm_bt needs M,N % 128 == 0 (pad host-side); got ",
                M, "x", N, "x", K);
    const int nwg = (M / 128) * (N / 128);
    gemm_bt_bf16_kernel<<<dim3(nwg), dim3(256), 0, cur_stream()>>>(
        bf16_ptr(A), bf16_ptr(B), bf16_mut(C), M, N, K);
  }
  HIP_CHECK_KERNEL();
  return C;
}

// ---------------- Grouped GEMM (MoE) ----------------
torch::Tensor grouped_gemm_bt(torch::Tensor A, torch::Tensor W,
                              torch::Tensor tile_expert, torch::Tensor tile_m0,
                              torch::Tensor seg_ends) {
  check_bf16(A, "A");
  check_bf16(W, "W");
  TORCH_CHECK(tile_expert.scalar_type() == torch::kInt32 && tile_expert.is_contiguous());
  TORCH_CHECK(tile_m0.scalar_type() == torch::kInt32 && tile_m0.is_contiguous());
  TORCH_CHECK(seg_ends.scalar_type() == torch::kInt32 && seg_ends.is_contiguous());
  const int K = A.size(1);
  const int N = W.size(1);
  TORCH_CHECK(W.size(2) == K && N % 128 == 0 && K % 64 == 0);
  const int n_mtiles = tile_expert.size(0);
  const int tiles_n = N / 128;
  auto C = torch::empty({A.size(0), (long)N}, A.options());
  if (n_mtiles > 0) {
    grouped_gemm_bt_bf16_kernel<<<dim3(n_mtiles * tiles_n), dim3(256), 0, cur_stream()>>>(
        bf16_ptr(A), bf16_ptr(W), bf16_mut(C), tile_expert.data_ptr<int>(),
        tile_m0.data_ptr<int>(), seg_ends.data_ptr<int>(), N, K, tiles_n);
    HIP_CHECK_KERNEL();
  }
  return C;
}

// ---------------- fp8 path ----------------
std::vector<torch::Tensor> quant_fp8(torch::Tensor x) {
  check_bf16(x, "x");
  const int K = x.size(-1);
  TORCH_CHECK(K % 8 == 0);
  const long long rows = x.numel() / K;
  auto q = torch::empty({rows, (long)K}, x.options().dtype(torch::kUInt8));
  auto s = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  quant_fp8_rowwise_kernel<<<dim3((unsigned)rows), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(x), q.data_ptr<unsigned char>(), s.data_ptr<float>(), K);
  HIP_CHECK_KERNEL();
  return {q, s};
}

torch::Tensor gemm_bt_fp8(torch::Tensor Aq, torch::Tensor As,
                          torch::Tensor Bq, torch::Tensor Bs) {
  TORCH_CHECK(Aq.scalar_type() == torch::kUInt8 && Bq.scalar_type() == torch::kUInt8);
  TORCH_CHECK(As.scalar_type() == torch::kFloat32 && Bs.scalar_type() == torch::kFloat32);
  const int M = Aq.size(0), K = Aq.size(1), N = Bq.size(0);
  TORCH_CHECK(Bq.size(1) == K && M % 128 == 0 && N % 128 == 0 && K % 128 == 0,
              "fp8 gemm needs M,N%128, K%128; got ", M, "x", N, "x", K);
  auto C = torch::empty({M, N}, Aq.options().dtype(torch::kBFloat16));
  const int nwg = (M / 128) * (N / 128);
  gemm_bt_fp8_kernel<<<dim3(nwg), dim3(256), 0, cur_stream()>>>(
      Aq.data_ptr<unsigned char>(), As.data_ptr<float>(),
      Bq.data_ptr<unsigned char>(), Bs.data_ptr<float>(),
      bf16_mut(C), M, N, K);
  HIP_CHECK_KERNEL();
  return C;
}

// ---------------- MX block-scaled fp8 path ----------------
std::vector<torch::Tensor> quant_mxfp8(torch::Tensor x) {
  check_bf16(x, "x");
  const int K = x.size(-1);
  TORCH_CHECK(K % 32 == 0, "mxfp8 needs K%32; got ", K);
  const long long rows = x.numel() / K;
  auto q = torch::empty({rows, (long)K}, x.options().dtype(torch::kUInt8));
  auto s = torch::empty({rows, (long)(K / 32)}, x.options().dtype(torch::kUInt8));
  quant_mxfp8_kernel<<<dim3((unsigned)rows), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(x), q.data_ptr<unsigned char>(), s.data_ptr<unsigned char>(), K);
  HIP_CHECK_KERNEL();
  return {q, s};
}

torch::Tensor gemm_bt_mxfp8(torch::Tensor Aq, torch::Tensor As,
                            torch::Tensor Bq, torch::Tensor Bs) {
  TORCH_CHECK(Aq.scalar_type() == torch::kUInt8 && Bq.scalar_type() == torch::kUInt8);
  TORCH_CHECK(As.scalar_type() == torch::kUInt8 && Bs.scalar_type() == torch::kUInt8);
  const int M = Aq.size(0), K = Aq.size(1), N = Bq.size(0);
  TORCH_CHECK(Bq.size(1) == K && M % 128 == 0 && N % 128 == 0 && K % 128 == 0,
              "mxfp8 gemm needs M,N%128, K%128; got ", M, "x", N, "x", K);
  TORCH_CHECK(As.size(1) == K / 32 && Bs.size(1) == K / 32);
  auto C = torch::empty({M, N}, Aq.options().dtype(torch::kBFloat16));
  const int nwg256 = (M % 256 == 0 && N % 256 == 0) ? (M / 256) * (N / 256) : 0;
  if (nwg256 >= 160) {
    // NOTE: a single-barrier pipeline port (gemm_bt_mxfp8_pipe_kernel, kept
    // for reference) measured SLOWER (918-998 TF/s vs 1187-1325): the
    // 32x32x64 f32x16 accumulators fragment the register budget at both
    // 512-thr (72 B/lane spill) and 1024-thr (176 B/lane) geometries.
    // Measured-and-documented rejection; see profiles/r02_gemm_pipeline.txt.
    gemm_bt_mxfp8_256_kernel<<<dim3(nwg256), dim3(512), 0, cur_stream()>>>(
        Aq.data_ptr<unsigned char>(), As.data_ptr<unsigned char>(),
        Bq.data_ptr<unsigned char>(), Bs.data_ptr<unsigned char>(),
        bf16_mut(C), M, N, K);
  } else {
    const int nwg = (M / 128) * (N / 128);
    gemm_bt_mxfp8_kernel<<<dim3(nwg), dim3(256), 0, cur_stream()>>>(
        Aq.data_ptr<unsigned char>(), As.data_ptr<unsigned char>(),
        Bq.data_ptr<unsigned char>(), Bs.data_ptr<unsigned char>(),
        bf16_mut(C), M, N, K);
  }
  HIP_CHECK_KERNEL();
  return C;
}

// decode fused add+RMSNorm+GEMV: returns (C [M,N], res_out=x(+res) [M,K])
std::vector<torch::Tensor> gemv_norm_bt(torch::Tensor x, c10::optional<torch::Tensor> res_in,
                                        torch::Tensor normw, torch::Tensor B, double eps) {
  check_bf16(x, "x");
  check_bf16(normw, "normw");
  check_bf16(B, "B");
  const int M = x.size(0), K = x.size(1), N = B.size(0);
  TORCH_CHECK(M <= 16 && N % 4 == 0 && K % 512 == 0 && B.size(1) == K);
  auto C = torch::empty({M, N}, x.options());
  auto res_out = torch::empty({M, K}, x.options());
  int MM = M <= 1 ? 1 : M <= 2 ? 2 : M <= 4 ? 4 : M <= 8 ? 8 : 16;
  torch::Tensor xp = x, rp;
  const ushort* rptr = nullptr;
  if (res_in.has_value()) {
    check_bf16(*res_in, "res_in");
    rp = *res_in;
  }
  if (M < MM) {
    xp = torch::zeros({MM, (long)K}, x.options());
    xp.narrow(0, 0, M).copy_(x);
    if (res_in.has_value()) {
      auto r2 = torch::zeros({MM, (long)K}, x.options());
      r2.narrow(0, 0, M).copy_(rp);
      rp = r2;
    }
  }
  if (res_in.has_value()) rptr = bf16_ptr(rp);
  const dim3 grid((N + 3) / 4);
  auto launch = [&](auto kern) {
    kern<<<grid, dim3(256), 0, cur_stream()>>>(bf16_ptr(xp), rptr,
                                               bf16_ptr(normw), bf16_ptr(B),
                                               bf16_mut(C), bf16_mut(res_out),
                                               M, N, K, (float)eps);
  };
  switch (MM) {
    case 1: launch(gemv_norm_bt_bf16_m1); break;
    case 2: launch(gemv_norm_bt_bf16_m2); break;
    case 4: launch(gemv_norm_bt_bf16_m4); break;
    case 8: launch(gemv_norm_bt_bf16_m8); break;
    default: launch(gemv_norm_bt_bf16_m16); break;
  }
  HIP_CHECK_KERNEL();
  return {C, res_out};
}

// ---------------- Flash attention prefill ----------------
torch::Tensor attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor vt,
                       double scale) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  check_bf16(vt, "vt");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hk = k.size(1);
  TORCH_CHECK(D == 128, "attn_fwd supports D=128");
  TORCH_CHECK(S % 64 == 0, "S must be padded to 64");
  TORCH_CHECK(vt.size(1) == Hk && vt.size(2) == D && vt.size(3) == S, "vt must be [B,Hk,D,S]");
  TORCH_CHECK(H % Hk == 0);
  auto o = torch::empty_like(q);
  attn_fwd_bf16_kernel<<<dim3(S / 64, H, B), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(q), bf16_ptr(k), bf16_ptr(vt), bf16_mut(o), B, H, Hk, S, (float)scale);
  HIP_CHECK_KERNEL();
  return o;
}

// v2: swapped-QK^T in-register-softmax kernel; returns O^T [B,H,D,S]
torch::Tensor attn_fwd_v2(torch::Tensor q, torch::Tensor k, torch::Tensor vt,
                          double scale) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  check_bf16(vt, "vt");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hk = k.size(1);
  TORCH_CHECK(D == 128 && S % 128 == 0 && H % Hk == 0);
  TORCH_CHECK(vt.size(1) == Hk && vt.size(2) == D && vt.size(3) == S);
  auto ot = torch::empty({B, H, D, S}, q.options());
  attn_fwd_v2_kernel<<<dim3(S / 128, H, B), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(q), bf16_ptr(k), bf16_ptr(vt), bf16_mut(ot), B, H, Hk, S, (float)scale);
  HIP_CHECK_KERNEL();
  return ot;
}

// v3: v2 structure with 8 waves sharing one staged K/V tile (256 q rows)
torch::Tensor attn_fwd_v3(torch::Tensor q, torch::Tensor k, torch::Tensor vt,
                          double scale) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  check_bf16(vt, "vt");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hk = k.size(1);
  TORCH_CHECK(D == 128 && S % 128 == 0 && H % Hk == 0);
  TORCH_CHECK(vt.size(1) == Hk && vt.size(2) == D && vt.size(3) == S);
  auto ot = torch::empty({B, H, D, S}, q.options());
  attn_fwd_v3_kernel<<<dim3((S + 255) / 256, H, B), dim3(512), 0, cur_stream()>>>(
      bf16_ptr(q), bf16_ptr(k), bf16_ptr(vt), bf16_mut(ot), B, H, Hk, S, (float)scale);
  HIP_CHECK_KERNEL();
  return ot;
}

// v4 probe family: v3 + {defer-max, async-STAGE register staging,
// glds-after-QK} lever matrix (attention_v4.hip header for the map)
#define DECL_V4(VAR)                                                         \
  extern "C" __global__ void attn_fwd_v4_##VAR##_kernel(                     \
      const ushort*, const ushort*, const ushort*, ushort*, int, int, int,   \
      int, float);
DECL_V4(0) DECL_V4(1) DECL_V4(2) DECL_V4(3)
DECL_V4(4) DECL_V4(5) DECL_V4(6) DECL_V4(7)
DECL_V4(8) DECL_V4(9) DECL_V4(10) DECL_V4(11) DECL_V4(12) DECL_V4(13) DECL_V4(14) DECL_V4(15) DECL_V4(16) DECL_V4(17) DECL_V4(18) DECL_V4(19)

torch::Tensor attn_fwd_v4(torch::Tensor q, torch::Tensor k, torch::Tensor vt,
                          double scale, long var) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  check_bf16(vt, "vt");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hk = k.size(1);
  TORCH_CHECK(D == 128 && S % 128 == 0 && H % Hk == 0);
  TORCH_CHECK(vt.size(1) == Hk && vt.size(2) == D && vt.size(3) == S);
  auto ot = torch::empty({B, H, D, S}, q.options());
  const bool four_wave = (var == 16 || var == 18);  // QBLK 128, 256 threads
  dim3 grid(four_wave ? (S + 127) / 128 : (S + 255) / 256, H, B);
  if (var == 19) {  // persistent: one block per CU walks the work items
    const int NW = ((S + 255) / 256) * H * B;
    grid = dim3(NW < 256 ? NW : 256, 1, 1);
  }
  const dim3 blk(four_wave ? 256 : 512);
  auto s = cur_stream();
#define LAUNCH_V4(VAR)                                                       \
  case VAR:                                                                  \
    attn_fwd_v4_##VAR##_kernel<<<grid, blk, 0, s>>>(                         \
        bf16_ptr(q), bf16_ptr(k), bf16_ptr(vt), bf16_mut(ot), B, H, Hk, S,  \
        (float)scale);                                                       \
    break;
  switch (var) {
    LAUNCH_V4(0) LAUNCH_V4(1) LAUNCH_V4(2) LAUNCH_V4(3)
    LAUNCH_V4(4) LAUNCH_V4(5) LAUNCH_V4(6) LAUNCH_V4(7)
    LAUNCH_V4(8) LAUNCH_V4(9) LAUNCH_V4(10) LAUNCH_V4(11) LAUNCH_V4(12) LAUNCH_V4(13) LAUNCH_V4(14) LAUNCH_V4(15) LAUNCH_V4(16) LAUNCH_V4(17) LAUNCH_V4(18) LAUNCH_V4(19)
    default: TORCH_CHECK(false, "attn_fwd_v4: unknown variant ", var);
  }
#undef LAUNCH_V4
  HIP_CHECK_KERNEL();
  return ot;
}

// v5 = promoted winner of the v4 lever matrix (variant 15: register-staged
// async prefetch, double LDS buffer, defer-max, softmax VALU diet, sm-split;
// 501 TF @ B4 H32 S2048 / 580 TF @ B1 H32 S8192 causal vs v3's 434/477 —
// profiles/r02_attn_ladder.txt)
torch::Tensor attn_fwd_v5(torch::Tensor q, torch::Tensor k, torch::Tensor vt,
                          double scale) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  check_bf16(vt, "vt");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hk = k.size(1);
  TORCH_CHECK(D == 128 && S % 128 == 0 && H % Hk == 0);
  TORCH_CHECK(vt.size(1) == Hk && vt.size(2) == D && vt.size(3) == S);
  auto ot = torch::empty({B, H, D, S}, q.options());
  attn_fwd_v4_15_kernel<<<dim3((S + 255) / 256, H, B), dim3(512), 0, cur_stream()>>>(
      bf16_ptr(q), bf16_ptr(k), bf16_ptr(vt), bf16_mut(ot), B, H, Hk, S, (float)scale);
  HIP_CHECK_KERNEL();
  return ot;
}

// ---------------- Paged decode attention ----------------
torch::Tensor paged_decode_attn(torch::Tensor q, torch::Tensor kcache,
                                torch::Tensor vcache, torch::Tensor block_table,
                                torch::Tensor ctx_lens, double scale) {
  check_bf16(q, "q");
  check_bf16(kcache, "kcache");
  check_bf16(vcache, "vcache");
  TORCH_CHECK(block_table.scalar_type() == torch::kInt32 && block_table.is_contiguous());
  TORCH_CHECK(ctx_lens.scalar_type() == torch::kInt32 && ctx_lens.is_contiguous());
  const int B = q.size(0), H = q.size(1), D = q.size(2);
  const int Hk = kcache.size(2);
  TORCH_CHECK(D == 128 && kcache.size(3) == D);
  TORCH_CHECK(kcache.size(1) == 16, "PAGE_SIZE=16");
  const int max_pages = block_table.size(1);
  auto o = torch::empty_like(q);
  // flash-decoding split-8: partial kernel fills the chip (H*B*8 blocks),
  // merge kernel combines slices.  Both graph-capturable.
  auto partials = torch::empty({B, H, 8, 2 + 128},
                               q.options().dtype(torch::kFloat32));
  paged_decode_attn_partial_kernel<<<dim3(H, B, 8), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(q), bf16_ptr(kcache), bf16_ptr(vcache),
      partials.data_ptr<float>(),
      block_table.data_ptr<int>(), ctx_lens.data_ptr<int>(), H, Hk, max_pages,
      (float)scale);
  paged_decode_attn_merge_kernel<<<dim3(H, B), dim3(64), 0, cur_stream()>>>(
      partials.data_ptr<float>(), bf16_mut(o), H);
  HIP_CHECK_KERNEL();
  return o;
}

// ---------------- Logit post-processing ----------------
torch::Tensor argmax_rows(torch::Tensor logits) {
  check_bf16(logits, "logits");
  const int rows = logits.size(0), vocab = logits.size(1);
  auto out = torch::empty({rows}, logits.options().dtype(torch::kInt32));
  argmax_rows_kernel<<<dim3(rows), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(logits), out.data_ptr<int>(), vocab);
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor target_logprob(torch::Tensor logits, torch::Tensor targets) {
  check_bf16(logits, "logits");
  TORCH_CHECK(targets.scalar_type() == torch::kInt32 && targets.is_contiguous());
  const int rows = logits.size(0), vocab = logits.size(1);
  TORCH_CHECK(targets.size(0) == rows);
  auto out = torch::empty({rows}, logits.options().dtype(torch::kFloat32));
  target_logprob_kernel<<<dim3(rows), dim3(256), 0, cur_stream()>>>(
      bf16_ptr(logits), targets.data_ptr<int>(), out.data_ptr<float>(), vocab);
  HIP_CHECK_KERNEL();
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm fwd (bf16)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm, "residual += x; y = rmsnorm(residual)");
  m.def("rope_inplace", &rope_inplace, "RoPE in place over q,k");
  m.def("rope_scatter_qkv", &rope_scatter_qkv, "fused RoPE + [B,H,S,D] scatter from qkv");
  m.def("vt_from_qkv", &vt_from_qkv, "LDS-tiled V^T [B,Hk,D,S] from the qkv slice");
  m.def("rope_kv_append", &rope_kv_append, "decode: slice+RoPE+KV-append in one kernel");
  m.def("gemv_norm_bt", &gemv_norm_bt, "decode: fused add+RMSNorm+GEMV");
  m.def("swiglu", &swiglu, "silu(gate)*up from fused gateup");
  m.def("add_bf16", &add_bf16, "a + b (bf16)");
  m.def("gemm_bt", &gemm_bt, "C = A @ B^T (bf16 MFMA)");
  m.def("gemm_bt_8ph", [](torch::Tensor a, torch::Tensor b) {
    check_bf16(a, "a"); check_bf16(b, "b");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 128 == 0);
    auto C = torch::empty({M, N}, a.options());
    gemm_bt_bf16_8ph_kernel<<<dim3((M / 256) * (N / 256)), dim3(512), 0, cur_stream()>>>(
        bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K);
    HIP_CHECK_KERNEL();
    return C;
  }, "256-tile 8-phase deep-pipelined GEMM (counted vmcnt)");
  m.def("gemm_bt_8ph_v", [](torch::Tensor a, torch::Tensor b, long var) {
    check_bf16(a, "a"); check_bf16(b, "b");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 128 == 0);
    auto C = torch::empty({M, N}, a.options());
    const dim3 g((M / 256) * (N / 256)), blk(512);
    switch (var) {
      case 1: gemm_bt_bf16_8ph_v1_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 2: gemm_bt_bf16_8ph_v2_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 3: gemm_bt_bf16_8ph_v3_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 4: gemm_bt_bf16_8ph_v4_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 5: gemm_bt_bf16_8ph_v5_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 6: gemm_bt_bf16_8ph_v6_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 7: gemm_bt_bf16_8ph_v7_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 8: gemm_bt_bf16_8ph_v8_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 9: gemm_bt_bf16_8ph_v9_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 11: gemm_bt_bf16_8ph_v11_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 12: gemm_bt_bf16_8ph_v12_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 13: gemm_bt_bf16_8ph_v13_kernel<<<dim3((M / 256) * (N / 256)), dim3(1024), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 14: gemm_bt_bf16_8ph_v14_kernel<<<dim3((M / 256) * (N / 256)), dim3(1024), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 16: gemm_bt_bf16_8ph_v16_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 17: gemm_bt_bf16_8ph_v17_kernel<<<dim3((M / 256) * (N / 128)), dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 18: gemm_bt_bf16_asm_kernel<<<g, dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 19: gemm_bt_bf16_asm2_kernel<<<g, dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 20: gemm_bt_bf16_asm3_kernel<<<g, dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 21: gemm_bt_bf16_asm4_kernel<<<g, dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 22: gemm_bt_bf16_asm5_kernel<<<g, dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 23: gemm_bt_bf16_asm6_kernel<<<g, dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      case 24: gemm_bt_bf16_asm7_kernel<<<g, dim3(512), 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
      default: gemm_bt_bf16_8ph_kernel<<<g, blk, 0, cur_stream()>>>(
                  bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K); break;
    }
    HIP_CHECK_KERNEL();
    return C;
  }, "8-phase GEMM structural variants (A/B probe)");
  m.def("gemv_bt_v3", [](torch::Tensor a, torch::Tensor b) {
    check_bf16(a, "a"); check_bf16(b, "b");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(M == 1 && K % 512 == 0 && K <= 14336 && N % 8 == 0);
    auto C = torch::empty({M, N}, a.options());
    if (K <= 4096)
      gemv_bt_bf16_v3_m1<<<dim3(N / 8), dim3(256), 0, cur_stream()>>>(
          bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K);
    else
      gemv_bt_bf16_v3w_m1<<<dim3(N / 8), dim3(256), 0, cur_stream()>>>(
          bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K);
    HIP_CHECK_KERNEL();
    return C;
  }, "loader/consumer LDS-DMA streaming GEMV (M=1)");
  m.def("gemv_bt_fp8w", [](torch::Tensor x, torch::Tensor bq, torch::Tensor bs) {
    check_bf16(x, "x");
    TORCH_CHECK(bq.scalar_type() == torch::kUInt8 && bs.scalar_type() == torch::kFloat32);
    const int M = x.size(0), K = x.size(1), N = bq.size(0);
    TORCH_CHECK(M >= 1 && M <= 8 && K % 1024 == 0 && bq.size(1) == K && N % 4 == 0);
    const int MM = M <= 1 ? 1 : M <= 2 ? 2 : M <= 4 ? 4 : 8;
    torch::Tensor xp = x;
    if (M < MM) {
      xp = torch::zeros({MM, (long)K}, x.options());
      xp.narrow(0, 0, M).copy_(x);
    }
    auto C = torch::empty({M, N}, x.options());
    // MM==1 runs 1 row/wave (grid N/4); MM>1 runs 2 rows/wave (N/8)
    const int nb = MM > 1 ? (N + 7) / 8 : (N + 3) / 4;
    auto launch = [&](auto kern) {
      kern<<<dim3(nb), dim3(256), 0, cur_stream()>>>(
          bf16_ptr(xp), bq.data_ptr<unsigned char>(), bs.data_ptr<float>(),
          bf16_mut(C), M, N, K);
    };
    switch (MM) {
      case 1: launch(gemv_bt_fp8w_m1); break;
      case 2: launch(gemv_bt_fp8w_m2); break;
      case 4: launch(gemv_bt_fp8w_m4); break;
      default: launch(gemv_bt_fp8w_m8); break;
    }
    HIP_CHECK_KERNEL();
    return C;
  }, "fp8-weight x bf16-activation GEMV (decode, M<=8)");
  m.def("gemv_bt_mxfp8w", [](torch::Tensor x, torch::Tensor bq, torch::Tensor bs) {
    check_bf16(x, "x");
    TORCH_CHECK(bq.scalar_type() == torch::kUInt8 && bs.scalar_type() == torch::kUInt8);
    const int M = x.size(0), K = x.size(1), N = bq.size(0);
    TORCH_CHECK(M >= 1 && M <= 8 && K % 1024 == 0 && bq.size(1) == K &&
                bs.size(1) == K / 32 && N % 4 == 0);
    const int MM = M <= 1 ? 1 : M <= 2 ? 2 : M <= 4 ? 4 : 8;
    torch::Tensor xp = x;
    if (M < MM) {
      xp = torch::zeros({MM, (long)K}, x.options());
      xp.narrow(0, 0, M).copy_(x);
    }
    auto C = torch::empty({M, N}, x.options());
    auto launch = [&](auto kern) {
      kern<<<dim3(N / 4), dim3(256), 0, cur_stream()>>>(
          bf16_ptr(xp), bq.data_ptr<unsigned char>(),
          bs.data_ptr<unsigned char>(), bf16_mut(C), M, N, K);
    };
    switch (MM) {
      case 1: launch(gemv_bt_mxfp8w_m1); break;
      case 2: launch(gemv_bt_mxfp8w_m2); break;
      case 4: launch(gemv_bt_mxfp8w_m4); break;
      default: launch(gemv_bt_mxfp8w_m8); break;
    }
    HIP_CHECK_KERNEL();
    return C;
  }, "mxfp8-weight x bf16-activation GEMV (decode, M<=8)");
  m.def("swiglu_gemv_bt", [](torch::Tensor gu, torch::Tensor b) {
    check_bf16(gu, "gateup"); check_bf16(b, "w");
    const int M = gu.size(0), K2 = gu.size(1), N = b.size(0);
    const int K = K2 / 2;
    TORCH_CHECK(M == 1 && K2 % 2 == 0 && K % 1024 == 0 && K > 4096 &&
                K <= 14336 && N % 8 == 0 && b.size(1) == K);
    auto C = torch::empty({M, N}, gu.options());
    swiglu_gemv_bt_bf16_m1<<<dim3(N / 8), dim3(256), 0, cur_stream()>>>(
        bf16_ptr(gu), bf16_ptr(b), bf16_mut(C), M, N, K);
    HIP_CHECK_KERNEL();
    return C;
  }, "fused silu(g)*u + streaming GEMV (decode down-proj, M=1)");
  m.def("gemm_bt_256x32", [](torch::Tensor a, torch::Tensor b) {
    check_bf16(a, "a"); check_bf16(b, "b");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0);
    auto C = torch::empty({M, N}, a.options());
    gemm_bt_bf16_256x32_kernel<<<dim3((M / 256) * (N / 256)), dim3(512), 0, cur_stream()>>>(
        bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K);
    HIP_CHECK_KERNEL();
    return C;
  }, "256-tile GEMM on 32x32x16 MFMA (A/B experiment)");
  m.def("gemm_bt_256sg", [](torch::Tensor a, torch::Tensor b) {
    check_bf16(a, "a"); check_bf16(b, "b");
    const int M = a.size(0), K = a.size(1), N = b.size(0);
    TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0);
    auto C = torch::empty({M, N}, a.options());
    gemm_bt_bf16_256sg_kernel<<<dim3((M / 256) * (N / 256)), dim3(512), 0, cur_stream()>>>(
        bf16_ptr(a), bf16_ptr(b), bf16_mut(C), M, N, K);
    HIP_CHECK_KERNEL();
    return C;
  }, "256-tile GEMM with sched_group_barrier interleave (A/B experiment)");
  m.def("grouped_gemm_bt", &grouped_gemm_bt, "segment-grouped C = A @ W[e]^T (MoE)");
  m.def("moe_combine", [](torch::Tensor down, torch::Tensor pos, torch::Tensor weight) {
    check_bf16(down, "down");
    TORCH_CHECK(pos.scalar_type() == torch::kInt32 && pos.dim() == 2);
    TORCH_CHECK(weight.scalar_type() == torch::kFloat32);
    const long long T = pos.size(0);
    const int k = pos.size(1), H = down.size(1);
    TORCH_CHECK(H % 8 == 0);
    auto out = torch::empty({T, (long)H}, down.options());
    moe_combine_kernel<<<dim3((unsigned)T), dim3(256), 0, cur_stream()>>>(
        bf16_ptr(down), pos.data_ptr<int>(), weight.data_ptr<float>(),
        bf16_mut(out), H, k);
    HIP_CHECK_KERNEL();
    return out;
  }, "routed-FFN weighted combine via inverse permutation");
  m.def("quant_fp8", &quant_fp8, "row-wise bf16 -> e4m3 + scale");
  m.def("gemm_bt_fp8", &gemm_bt_fp8, "fp8 MFMA GEMM with row/col rescale");
  m.def("quant_mxfp8", &quant_mxfp8, "OCP MX quant: bf16 -> e4m3 + e8m0 per-32 scales");
  m.def("gemm_bt_mxfp8", &gemm_bt_mxfp8, "MX block-scaled fp8 MFMA GEMM (32x32x64)");
  m.def("attn_fwd", &attn_fwd, "causal flash attention fwd (D=128, GQA)");
  m.def("attn_fwd_v2", &attn_fwd_v2, "swapped-QK^T attention; O^T out [B,H,D,S]");
  m.def("attn_fwd_v3", &attn_fwd_v3, "v2 with 8-wave shared K/V tiles; O^T out");
  m.def("attn_fwd_v4", &attn_fwd_v4, "v4 lever-matrix probe (variant arg)");
  m.def("attn_fwd_v5", &attn_fwd_v5, "production attention (v4 winner); O^T out");
  m.def("paged_decode_attn", &paged_decode_attn, "paged decode attention");
  m.def("argmax_rows", &argmax_rows, "row argmax over bf16 logits");
  m.def("target_logprob", &target_logprob, "fused log_softmax gather");
}
