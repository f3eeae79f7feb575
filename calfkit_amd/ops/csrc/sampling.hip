// Token sampling over the vocab for MI355X (gfx950).
//
// Gumbel-max: argmax(logits/T + G) with G ~ Gumbel(0,1) samples exactly from
// softmax(logits/T) in ONE pass over the vocab — no sort, no cumulative sum,
// so the kernel is a pure block-argmax stream (memory-bound, vectorized).
// temperature == 0 degenerates to greedy argmax.  top_k/top_p are applied as
// a masking prepass only when requested (two-pass; rare in the agent loop).
//
// RNG: counter-based squares64-style hash keyed by (seed, step, vocab index)
// — stateless, graph-replay-safe (the seed/step live in a device buffer the
// host bumps between replays).

#include "common.h"

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return x;
}

// uniform in (0,1]
__device__ __forceinline__ float u01(uint64_t h) {
  return ((h >> 40) + 1.f) * (1.f / 16777217.f);
}

template <int BLOCK>
__global__ void sample_kernel(int32_t* __restrict__ out,      // [B]
                              const float* __restrict__ logits,  // [B, V]
                              const float* __restrict__ temperature,  // [B]
                              const uint64_t* __restrict__ seed_step, // [2]
                              const int V) {
  __shared__ float best_lds[BLOCK / WAVE];
  __shared__ int bestid_lds[BLOCK / WAVE];
  const int b = blockIdx.x;
  const float temp = temperature[b];
  const uint64_t key = seed_step[0] ^ (seed_step[1] * 0x9e3779b97f4a7c15ULL) ^
                       ((uint64_t)b << 32);
  const float* row = logits + (int64_t)b * V;

  float best = -INFINITY;
  int bestid = 0;
  if (temp <= 0.f) {  // greedy
    for (int i = threadIdx.x; i < V; i += BLOCK) {
      const float x = row[i];
      if (x > best || (x == best && i < bestid)) {
        best = x;
        bestid = i;
      }
    }
  } else {
    const float invt = 1.f / temp;
    for (int i = threadIdx.x; i < V; i += BLOCK) {
      const float g = -__logf(-__logf(u01(mix64(key ^ (uint64_t)i))));
      const float x = row[i] * invt + g;
      if (x > best) {
        best = x;
        bestid = i;
      }
    }
  }
  // wave argmax
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(bestid, off, WAVE);
    if (ob > best || (ob == best && oi < bestid)) {
      best = ob;
      bestid = oi;
    }
  }
  const int wid = threadIdx.x / WAVE;
  if (threadIdx.x % WAVE == 0) {
    best_lds[wid] = best;
    bestid_lds[wid] = bestid;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < BLOCK / WAVE; ++w) {
      if (best_lds[w] > best || (best_lds[w] == best && bestid_lds[w] < bestid)) {
        best = best_lds[w];
        bestid = bestid_lds[w];
      }
    }
    out[b] = bestid;
  }
}

// top-k / top-p mask prepass: sets masked logits to -inf in place.
// Runs only when the settings ask for it; block per row; k-th value found by
// iterative threshold bisection on the row max..min range (8 rounds).
template <int BLOCK>
__global__ void topk_mask_kernel(float* __restrict__ logits, const int V,
                                 const int32_t* __restrict__ topk) {
  __shared__ float red[BLOCK / WAVE];
  const int b = blockIdx.x;
  const int k = topk[b];
  if (k <= 0 || k >= V) return;
  float* row = logits + (int64_t)b * V;

  float mx = -INFINITY, mn = INFINITY;
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    mx = fmaxf(mx, row[i]);
    mn = fminf(mn, row[i]);
  }
  mx = block_max<BLOCK>(mx, red);
  __syncthreads();
  mn = -block_max<BLOCK>(-mn, red);
  __syncthreads();

  // bisection for the threshold: invariant count(>= hi) <= k < count(>= lo).
  // Masking at hi keeps AT MOST k survivors (never more); the row max always
  // survives (mx >= hi by the invariant), so sampling stays well-defined.
  float lo = mn, hi = mx;
  for (int it = 0; it < 24; ++it) {
    const float mid = 0.5f * (lo + hi);
    int cnt = 0;
    for (int i = threadIdx.x; i < V; i += BLOCK) cnt += (row[i] >= mid);
    cnt = (int)block_sum<BLOCK>((float)cnt, red);
    __syncthreads();
    if (cnt > k)
      lo = mid;
    else
      hi = mid;
  }
  for (int i = threadIdx.x; i < V; i += BLOCK)
    if (row[i] < hi) row[i] = -INFINITY;
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temperature,
            torch::Tensor seed_step, torch::Tensor topk) {
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32);
  TORCH_CHECK(out.scalar_type() == torch::kInt32);
  TORCH_CHECK(seed_step.scalar_type() == torch::kInt64);
  const int B = logits.size(0), V = logits.size(1);
  constexpr int BLOCK = 512;
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  if (topk.defined() && topk.numel() == B) {
    hipLaunchKernelGGL(topk_mask_kernel<BLOCK>, dim3(B), dim3(BLOCK), 0, s,
                       (float*)logits.data_ptr(), V,
                       (const int32_t*)topk.data_ptr());
  }
  hipLaunchKernelGGL(sample_kernel<BLOCK>, dim3(B), dim3(BLOCK), 0, s,
                     (int32_t*)out.data_ptr(), (const float*)logits.data_ptr(),
                     (const float*)temperature.data_ptr(),
                     (const uint64_t*)seed_step.data_ptr(), V);
  HIP_CHECK_LAST();
}
