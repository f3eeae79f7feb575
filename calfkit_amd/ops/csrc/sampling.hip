// Token sampling over the vocab for MI355X (gfx950).
//
// Gumbel-max: argmax(logits/T + G) with G ~ Gumbel(0,1) samples exactly
// from softmax(logits/T) in ONE pass over the vocab — no sort, no cumsum.
// temperature == 0 degenerates to greedy argmax.
//
// Two-stage argmax: a decode batch of 32 rows would occupy only 32 CUs with
// a block-per-row design (the r01 profile showed 83 us for a 16 MB scan);
// stage 1 fans (row, vocab-slice) pairs across the chip, stage 2 reduces
// the per-slice candidates.  Logits are consumed as bf16 directly — no
// fp32 materialization of [B, V] between the lm_head GEMM and sampling.
//
// RNG: counter-based mix64 keyed by (seed, step, vocab index) — stateless,
// graph-replay-safe (the seed/step pair lives in a device buffer).

#include "common.h"

#define NSLICE_MAX 64

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

template <typename T>
__device__ __forceinline__ float to_f32(T x);
template <>
__device__ __forceinline__ float to_f32<float>(float x) { return x; }
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 x) {
  return bf2f(x);
}

// stage 1: best (value, index) of one (row, slice)
template <typename T, int BLOCK>
__global__ void sample_partial_kernel(
    float2* __restrict__ cand,           // [B, NS] (best, bitcast index)
    const T* __restrict__ logits,        // [B, V]
    const float* __restrict__ temperature,
    const uint64_t* __restrict__ seed_step,
    const uint64_t* __restrict__ row_seeds,  // [B] or null: per-request keys
    const int V, const int ns) {
  __shared__ float best_lds[BLOCK / WAVE];
  __shared__ int bestid_lds[BLOCK / WAVE];
  const int b = blockIdx.x;
  const int sl = blockIdx.y;
  const float temp = temperature[b];
  // per-row key (request seed x token position, built host-side) makes the
  // draw independent of batch composition — same request seed => same
  // tokens regardless of what else is running.  The legacy
  // (seed, step, row) key remains for callers without row seeds.
  const uint64_t key =
      row_seeds ? mix64(row_seeds[b])
                : seed_step[0] ^ (seed_step[1] * 0x9e3779b97f4a7c15ULL) ^
                      ((uint64_t)b << 32);
  const T* row = logits + (int64_t)b * V;
  const int lo = (int)(((int64_t)V * sl) / ns);
  const int hi = (int)(((int64_t)V * (sl + 1)) / ns);

  float best = -INFINITY;
  int bestid = lo;
  if (temp <= 0.f) {
    for (int i = lo + threadIdx.x; i < hi; i += BLOCK) {
      const float x = to_f32(row[i]);
      if (x > best || (x == best && i < bestid)) {
        best = x;
        bestid = i;
      }
    }
  } else {
    const float invt = 1.f / temp;
    for (int i = lo + threadIdx.x; i < hi; i += BLOCK) {
      const float g = -__logf(-__logf(u01(mix64(key ^ (uint64_t)i))));
      const float x = to_f32(row[i]) * invt + g;
      if (x > best) {
        best = x;
        bestid = i;
      }
    }
  }
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
    for (int w = 1; w < BLOCK / WAVE; ++w)
      if (best_lds[w] > best ||
          (best_lds[w] == best && bestid_lds[w] < bestid)) {
        best = best_lds[w];
        bestid = bestid_lds[w];
      }
    cand[(int64_t)b * ns + sl] = make_float2(best, __int_as_float(bestid));
  }
}

// stage 2: one small block per row folds the slice candidates
__global__ void sample_final_kernel(int32_t* __restrict__ out,
                                    const float2* __restrict__ cand,
                                    const int ns) {
  const int b = blockIdx.x;
  float best = -INFINITY;
  int bestid = 0;
  for (int sl = threadIdx.x; sl < ns; sl += WAVE) {
    const float2 c = cand[(int64_t)b * ns + sl];
    const int id = __float_as_int(c.y);
    if (c.x > best || (c.x == best && id < bestid)) {
      best = c.x;
      bestid = id;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(bestid, off, WAVE);
    if (ob > best || (ob == best && oi < bestid)) {
      best = ob;
      bestid = oi;
    }
  }
  if (threadIdx.x == 0) out[b] = bestid;
}

// top-k mask prepass (opt-in): sets sub-threshold logits to -inf in place.
template <typename T, int BLOCK>
__global__ void topk_mask_kernel(T* __restrict__ logits, const int V,
                                 const int32_t* __restrict__ topk) {
  __shared__ float red[BLOCK / WAVE];
  const int b = blockIdx.x;
  const int k = topk[b];
  if (k <= 0 || k >= V) return;
  T* row = logits + (int64_t)b * V;

  float mx = -INFINITY, mn = INFINITY;
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    const float x = to_f32(row[i]);
    mx = fmaxf(mx, x);
    mn = fminf(mn, x);
  }
  mx = block_max<BLOCK>(mx, red);
  __syncthreads();
  mn = -block_max<BLOCK>(-mn, red);
  __syncthreads();

  // bisection: count(>= hi) <= k < count(>= lo); mask below hi keeps <= k
  float lo = mn, hi = mx;
  for (int it = 0; it < 24; ++it) {
    const float mid = 0.5f * (lo + hi);
    int cnt = 0;
    for (int i = threadIdx.x; i < V; i += BLOCK)
      cnt += (to_f32(row[i]) >= mid);
    cnt = (int)block_sum<BLOCK>((float)cnt, red);
    __syncthreads();
    if (cnt > k)
      lo = mid;
    else
      hi = mid;
  }
  for (int i = threadIdx.x; i < V; i += BLOCK)
    if (to_f32(row[i]) < hi) row[i] = (T)(-INFINITY);
}

// ---- tool-call grammar bias -------------------------------
//
// Constrained sampling that steers the (random-init) model through the
// SyntheticTokenizer's tool-call grammar:
//     ... free text ... TCALL <name> <arg x ARG_TOKS> TEND ... EOS
// The NAME is genuinely sampled on-device among the advertised tools'
// name-token ids; args are free samples the parser materializes against
// the tool's JSON schema.  State is a per-row i32 (low byte = state,
// high bits = calls made), transitioned on the PREVIOUS sampled token —
// graph-replay-safe: all inputs live in static device buffers and the
// host mirrors transitions after each run.
//
// States: 0 = free text; 1 = prev was TCALL (force a name);
//         2+k = name + k arg tokens emitted (k == ARG_TOKS forces TEND).

#define G_TCALL 5
#define G_TEND 6
#define G_EOS 2
#define G_ARG_TOKS 4
#define G_MAX_TOOLS 8
#define G_BIG 65536.f

template <typename T>
__global__ void grammar_bias_kernel(
    T* __restrict__ logits,              // [B, V]
    const int64_t* __restrict__ prev,    // [B] previous sampled token
    int32_t* __restrict__ gstate,        // [B] state | calls<<8
    const int32_t* __restrict__ trigger, // [B] abs position; <0 = off
    const int32_t* __restrict__ names,   // [B, G_MAX_TOOLS]; -1 padded
    const int32_t* __restrict__ pos,     // [B] current position
    const int32_t* __restrict__ ncalls,  // [B] calls budget
    const int B, const int V) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  if (trigger[b] < 0) return;            // no tool plan for this row
  int st = gstate[b];
  int s = st & 0xff, calls = st >> 8;
  const int p = (int)prev[b];
  // transition on the previous token
  if (s == 1) {
    s = 2;                               // prev was the name
  } else if (s >= 2) {
    if (p == G_TEND) { s = 0; calls += 1; }
    else s += 1;
  } else if (s == 0 && p == G_TCALL) {
    s = 1;
  }
  T* row = logits + (int64_t)b * V;
  const auto bump = [&](int t, float d) {
    row[t] = (T)(to_f32(row[t]) + d);
  };
  if (s == 0) {
    if (calls < ncalls[b] && pos[b] >= trigger[b]) {
      bump(G_TCALL, G_BIG);
    } else if (calls > 0 && calls >= ncalls[b]) {
      bump(G_EOS, G_BIG);
    } else {
      // before the trigger: no call yet, and no premature end of turn
      bump(G_TCALL, -G_BIG);
      bump(G_EOS, -G_BIG);
    }
  } else if (s == 1) {
    for (int i = 0; i < G_MAX_TOOLS; ++i) {
      const int t = names[(int64_t)b * G_MAX_TOOLS + i];
      if (t >= 0) bump(t, G_BIG);
    }
  } else if (s - 2 >= G_ARG_TOKS) {
    bump(G_TEND, G_BIG);                 // close the call
  } else {
    // free arg tokens; the whole reserved/special range stays out
    for (int t = 0; t < 16; ++t) bump(t, -G_BIG);
  }
  gstate[b] = s | (calls << 8);
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void grammar_bias(torch::Tensor logits, torch::Tensor prev,
                  torch::Tensor gstate, torch::Tensor trigger,
                  torch::Tensor names, torch::Tensor pos,
                  torch::Tensor ncalls) {
  const int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(prev.scalar_type() == torch::kInt64);
  TORCH_CHECK(gstate.scalar_type() == torch::kInt32);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  const int blocks = (B + 63) / 64;
#define G(T)                                                                 \
  hipLaunchKernelGGL((grammar_bias_kernel<T>), dim3(blocks), dim3(64), 0, s, \
                     (T*)logits.data_ptr(), (const int64_t*)prev.data_ptr(), \
                     (int32_t*)gstate.data_ptr(),                            \
                     (const int32_t*)trigger.data_ptr(),                     \
                     (const int32_t*)names.data_ptr(),                       \
                     (const int32_t*)pos.data_ptr(),                         \
                     (const int32_t*)ncalls.data_ptr(), B, V)
  if (logits.scalar_type() == torch::kBFloat16)
    G(__hip_bfloat16);
  else
    G(float);
#undef G
  HIP_CHECK_LAST();
}

void sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temperature,
            torch::Tensor seed_step, torch::Tensor topk,
            torch::Tensor cand, torch::Tensor row_seeds) {
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(out.scalar_type() == torch::kInt32);
  TORCH_CHECK(seed_step.scalar_type() == torch::kInt64);
  const int B = logits.size(0), V = logits.size(1);
  const uint64_t* rs = nullptr;
  if (row_seeds.defined() && row_seeds.numel() >= B) {
    TORCH_CHECK(row_seeds.scalar_type() == torch::kInt64);
    rs = (const uint64_t*)row_seeds.data_ptr();
  }
  // enough slices to fill the chip several blocks deep (the B=256 run at
  // ns=2 measured 100 us for a 65 MB scan — launch-breadth bound)
  int ns = 1;
  while (B * ns < 2048 && ns < NSLICE_MAX) ns *= 2;
  TORCH_CHECK(cand.numel() >= (int64_t)B * ns * 2, "cand scratch too small");
  constexpr int BLOCK = 256;
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  const bool bf16 = logits.scalar_type() == torch::kBFloat16;

  if (topk.defined() && topk.numel() == B) {
    if (bf16)
      hipLaunchKernelGGL((topk_mask_kernel<__hip_bfloat16, 512>), dim3(B),
                         dim3(512), 0, s,
                         (__hip_bfloat16*)logits.data_ptr(), V,
                         (const int32_t*)topk.data_ptr());
    else
      hipLaunchKernelGGL((topk_mask_kernel<float, 512>), dim3(B), dim3(512),
                         0, s, (float*)logits.data_ptr(), V,
                         (const int32_t*)topk.data_ptr());
  }
#define P(T)                                                                \
  hipLaunchKernelGGL((sample_partial_kernel<T, BLOCK>), dim3(B, ns),        \
                     dim3(BLOCK), 0, s, (float2*)cand.data_ptr(),           \
                     (const T*)logits.data_ptr(),                           \
                     (const float*)temperature.data_ptr(),                  \
                     (const uint64_t*)seed_step.data_ptr(), rs, V, ns)
  if (bf16)
    P(__hip_bfloat16);
  else
    P(float);
#undef P
  hipLaunchKernelGGL(sample_final_kernel, dim3(B), dim3(WAVE), 0, s,
                     (int32_t*)out.data_ptr(),
                     (const float2*)cand.data_ptr(), ns);
  HIP_CHECK_LAST();
}
