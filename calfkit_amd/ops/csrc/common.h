// Common device helpers for calfkit_amd CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA4 — every cross-lane idiom below assumes it.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE 64

// 16-byte vector of 8 bf16 values — the coalescing sweet spot (8-16 B/lane).
struct alignas(16) bf16x8 {
  __hip_bfloat16 v[8];
};

struct alignas(8) bf16x4 {
  __hip_bfloat16 v[4];
};

__device__ __forceinline__ float bf2f(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
__device__ __forceinline__ __hip_bfloat16 f2bf(float x) {
  return __float2bfloat16(x);
}

// Wave-wide reductions (64 lanes).
__device__ __forceinline__ float wave_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}
__device__ __forceinline__ float wave_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block reduction over up to 16 waves via LDS (caller provides the scratch).
template <int BLOCK>
__device__ __forceinline__ float block_sum(float x, float* lds) {
  constexpr int NW = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  x = wave_sum(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  float r = (lane < NW) ? lds[lane] : 0.f;
  r = wave_sum(r);
  return __shfl(r, 0, WAVE);
}

template <int BLOCK>
__device__ __forceinline__ float block_max(float x, float* lds) {
  constexpr int NW = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  x = wave_max(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  float r = (lane < NW) ? lds[lane] : -INFINITY;
  r = wave_max(r);
  return __shfl(r, 0, WAVE);
}

#define HIP_CHECK_LAST()                                    \
  do {                                                      \
    hipError_t e_ = hipGetLastError();                      \
    if (e_ != hipSuccess) {                                 \
      TORCH_CHECK(false, "HIP kernel launch failed: ",      \
                  hipGetErrorString(e_));                   \
    }                                                       \
  } while (0)
