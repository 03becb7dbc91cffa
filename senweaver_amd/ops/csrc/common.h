// Common helpers for the senweaver_amd CDNA4 (gfx950) kernel library.
// Target: MI355X only — wave64, MFMA, LDS 160 KiB/CU, no multi-arch dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

using bf16_t = __hip_bfloat16;

// 16-byte vector of 8 bf16 values (the coalescing sweet spot on CDNA4:
// 16 B/lane x 64 lanes = 1 KiB per instruction).
struct alignas(16) bf16x8 {
  ushort v[8];
};

struct alignas(8) bf16x4 {
  ushort v[4];
};

__device__ __forceinline__ float bf2f(ushort u) {
  union {
    unsigned int i;
    float f;
  } c;
  c.i = (unsigned int)u << 16;
  return c.f;
}

__device__ __forceinline__ ushort f2bf(float f) {
  union {
    unsigned int i;
    float f;
  } c;
  c.f = f;
  // round-to-nearest-even, matching PyTorch's float->bfloat16 conversion
  unsigned int lsb = (c.i >> 16) & 1u;
  unsigned int rounded = c.i + 0x7fffu + lsb;
  return (ushort)(rounded >> 16);
}

// Wave-level reductions (64-wide). All lanes receive the result.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Reduce within each contiguous group of N lanes (N power of two <= 64).
template <int N>
__device__ __forceinline__ float group_reduce_sum(float v) {
#pragma unroll
  for (int off = N / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

template <int N>
__device__ __forceinline__ float group_reduce_max(float v) {
#pragma unroll
  for (int off = N / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

#define HIP_CHECK_KERNEL()                                     \
  do {                                                         \
    hipError_t e = hipGetLastError();                          \
    if (e != hipSuccess) {                                     \
      TORCH_CHECK(false, "HIP kernel launch failed: ",         \
                  hipGetErrorString(e));                       \
    }                                                          \
  } while (0)
