// Logit post-processing kernels: row argmax (greedy decode), fused
// target-token log-prob (the beam scorer's hot reduction), and top-k logits.
//
// These replace a full softmax materialization over the 128k Llama vocab:
// per row we stream the logits once (memory-bound, vectorized bf16x8) and
// produce either the argmax or log_softmax(logits)[target] directly.

#include "common.h"

// ---------------------------------------------------------------------------
// Row argmax over bf16 logits [rows, vocab] -> int32 [rows].
// Ties resolve to the lowest index (matches torch.argmax on CPU/GPU).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
argmax_rows_kernel(const ushort* __restrict__ logits, int* __restrict__ out,
                   int vocab) {
  const long long base = (long long)blockIdx.x * vocab;
  float best = -INFINITY;
  int best_idx = 0x7fffffff;
  const int nvec = vocab / 8;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(logits + base + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v.v[j]);
      int idx = i * 8 + j;
      if (f > best || (f == best && idx < best_idx)) {
        best = f;
        best_idx = idx;
      }
    }
  }
  // tail (vocab not multiple of 8)
  for (int idx = nvec * 8 + threadIdx.x; idx < vocab; idx += blockDim.x) {
    float f = bf2f(logits[base + idx]);
    if (f > best || (f == best && idx < best_idx)) {
      best = f;
      best_idx = idx;
    }
  }
  // wave reduce (value, index)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(best_idx, off, 64);
    if (ov > best || (ov == best && oi < best_idx)) {
      best = ov;
      best_idx = oi;
    }
  }
  __shared__ float sv[4];
  __shared__ int si[4];
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) {
    sv[wid] = best;
    si[wid] = best_idx;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < 4; ++w) {
      if (sv[w] > best || (sv[w] == best && si[w] < best_idx)) {
        best = sv[w];
        best_idx = si[w];
      }
    }
    out[blockIdx.x] = best_idx;
  }
}

// ---------------------------------------------------------------------------
// Fused log-prob of a target token per row:
//   out[r] = logits[r][tgt[r]] - max_r - log(sum(exp(logits[r] - max_r)))
// Single streaming pass computes max and (shifted) sumexp online, so the
// 128k-vocab row is read exactly once.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
target_logprob_kernel(const ushort* __restrict__ logits,
                      const int* __restrict__ targets,
                      float* __restrict__ out, int vocab) {
  const long long base = (long long)blockIdx.x * vocab;
  float m = -INFINITY;  // running max
  float s = 0.f;        // running sum of exp(x - m)
  const int nvec = vocab / 8;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(logits + base + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v.v[j]);
      if (f > m) {
        s *= __expf(m - f);
        m = f;
      }
      s += __expf(f - m);
    }
  }
  for (int idx = nvec * 8 + threadIdx.x; idx < vocab; idx += blockDim.x) {
    float f = bf2f(logits[base + idx]);
    if (f > m) {
      s *= __expf(m - f);
      m = f;
    }
    s += __expf(f - m);
  }
  // combine (m, s) across lanes: s_total at m_max
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float om = __shfl_xor(m, off, 64);
    float os = __shfl_xor(s, off, 64);
    float nm = fmaxf(m, om);
    s = s * __expf(m - nm) + os * __expf(om - nm);
    m = nm;
  }
  __shared__ float sm[4], ssum[4];
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) {
    sm[wid] = m;
    ssum[wid] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = sm[0], S = ssum[0];
#pragma unroll
    for (int w = 1; w < 4; ++w) {
      float nm = fmaxf(M, sm[w]);
      S = S * __expf(M - nm) + ssum[w] * __expf(sm[w] - nm);
      M = nm;
    }
    const float tl = bf2f(logits[base + targets[blockIdx.x]]);
    out[blockIdx.x] = tl - M - __logf(S);
  }
}
