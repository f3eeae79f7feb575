// Grouped MoE decode GEMM for MI355X (gfx950).
//
// Capability parity: the reference's MoE expert-parallel GEMM path
// (SURVEY §2.4 "Grouped GEMM (MoE)") — here as a CDNA4 weight-streaming
// kernel instead of a CUTLASS grouped GEMM.
//
// Regime: Mixtral decode.  R = T*topk token-expert assignments are
// bucketed per expert (device-side argsort; offsets via cumsum).  Every
// active expert's weights are read exactly once per step, so — like the
// dense skinny GEMM this is modeled on (skinny_gemm.hip) — the kernel's
// job is to stream W at the HBM ceiling with the MFMA work riding along.
// The expert loop moves onto the GRID (blockIdx.y = expert), the token
// gather rides the A-fragment load (indirection through tok[]), and the
// router scale rides the epilogue — so the whole MoE layer runs without
// a host sync and is hipGraph-capturable (the host-looped fallback in
// models/mixtral.py syncs on counts.tolist() every layer).
//
// Decomposition:
//   grid = (N/64, E); 4 waves; each wave owns 1/4 of K (in-block split-K,
//   splitk always 1 — block count N/64*E far exceeds 256 CUs already).
//   Buckets larger than MT=64 rows loop in MT chunks (weights re-stream
//   per chunk; at the decode shapes this path serves, avg bucket size is
//   R/E <= 64, so the common case is one pass).
//
// Fragments (mfma_f32_16x16x32_bf16, lane maps verified by GPU tests):
//   A[i][k]: lane i=l&15, k=(l>>4)*8+m  -> x row tok[i], 16 B contiguous
//   B[k][j]: lane j=l&15, k=(l>>4)*8+m  -> W^T row n0+j, 16 B contiguous
//   C[r][c]: lane c=l&15, r=(l>>4)*4+reg

#include "common.h"

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;

#define MFMA16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

#define BN 64       // output columns per block
#define MGWAVES 4   // waves per block (in-block K split)
#define MOE_MT 64   // rows per M-chunk

// x rows are gathered through tok[] (nullptr = identity: rows are already
// bucket-ordered, the down projection's case).  direct=0 writes the fp32
// slab work[r, :N] (gate-up; the SiLU epilogue mixes columns); direct=1
// writes bf16 out[r, :N] scaled by scale[r] (down projection; the router
// weight fuses here so the scatter-add back to tokens is a plain
// index_add over pre-scaled rows).
// MT is the rows-per-chunk template: every chunk past the first re-streams
// the expert's weights, so MT=128 (accumulators for 8 m-tiles live across
// the whole k-loop) halves the re-read when buckets run big (decode batch
// 256 -> avg bucket 64, tails ~90).  The LDS reduce stays a 64-row panel
// (two passes at MT=128) so occupancy holds at 2 blocks/CU.
template <int KU, int MT>
__global__ __launch_bounds__(MGWAVES * WAVE) void moe_gemm_kernel(
    float* __restrict__ work,             // [R, N] fp32 slab (direct=0)
    __hip_bfloat16* __restrict__ out,     // [R, N] bf16 (direct=1)
    const __hip_bfloat16* __restrict__ x, // [T, K] source rows
    const __hip_bfloat16* __restrict__ w, // [E, N, K] stacked expert W^T
    const int* __restrict__ tok,          // [R] row gather index (or null)
    const int* __restrict__ off,          // [E+1] bucket offsets
    const float* __restrict__ scale,      // [R] row scale (or null)
    const int N, const int K, const int direct) {
  constexpr int MTILES = MT / 16;
  const int n0 = blockIdx.x * BN;
  const int e = blockIdx.y;
  const int m_begin = off[e], m_end = off[e + 1];
  if (m_begin >= m_end) return;

  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int l16 = lane & 15;
  const int l4 = lane >> 4;

  const __hip_bfloat16* we = w + (int64_t)e * N * K;
  const int wave_k = K / MGWAVES;
  const int k_begin = wid * wave_k;
  const int k_end = k_begin + wave_k;
  const int wrow = n0 + l16;  // W^T row for this lane's B fragments

  __shared__ float red[MGWAVES][MOE_MT][BN];  // 64-row panel, 64 KB

  for (int m0 = m_begin; m0 < m_end; m0 += MT) {
    f32x4 acc[MTILES][BN / 16];
#pragma unroll
    for (int mi = 0; mi < MTILES; ++mi)
#pragma unroll
      for (int ni = 0; ni < BN / 16; ++ni)
        acc[mi][ni] = (f32x4){0.f, 0.f, 0.f, 0.f};

    // A row per m-tile for this lane; clamp into the bucket, gather via tok
    int arow[MTILES];
#pragma unroll
    for (int mi = 0; mi < MTILES; ++mi) {
      const int r = min(m0 + mi * 16 + l16, m_end - 1);
      arow[mi] = tok ? tok[r] : r;
    }

    for (int k = k_begin; k < k_end; k += KU) {
      bf16x8v a[KU / 32][MTILES], b[KU / 32][BN / 16];
#pragma unroll
      for (int kk = 0; kk < KU; kk += 32) {
#pragma unroll
        for (int mi = 0; mi < MTILES; ++mi)
          a[kk / 32][mi] = *reinterpret_cast<const bf16x8v*>(
              x + (int64_t)arow[mi] * K + k + kk + l4 * 8);
#pragma unroll
        for (int ni = 0; ni < BN / 16; ++ni)
          b[kk / 32][ni] = *reinterpret_cast<const bf16x8v*>(
              we + (int64_t)(wrow + ni * 16) * K + k + kk + l4 * 8);
      }
#pragma unroll
      for (int kk = 0; kk < KU; kk += 32)
#pragma unroll
        for (int ni = 0; ni < BN / 16; ++ni)
#pragma unroll
          for (int mi = 0; mi < MTILES; ++mi)
            acc[mi][ni] = MFMA16(a[kk / 32][mi], b[kk / 32][ni], acc[mi][ni]);
    }

    // cross-wave reduce through LDS in 64-row panels (MT/64 passes), then
    // one guarded row-range write per panel.  NOTE the guard r < m_end is
    // load-bearing in slab mode: rows past this bucket belong to the NEXT
    // expert's blocks.
#pragma unroll
    for (int p = 0; p < MT / MOE_MT; ++p) {
      __syncthreads();  // red[] reuse across panels/m-chunks
#pragma unroll
      for (int mi = 0; mi < MOE_MT / 16; ++mi)
#pragma unroll
        for (int ni = 0; ni < BN / 16; ++ni)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            red[wid][mi * 16 + l4 * 4 + r][ni * 16 + l16] =
                acc[p * (MOE_MT / 16) + mi][ni][r];
      __syncthreads();

      for (int i = threadIdx.x; i < MOE_MT * BN; i += MGWAVES * WAVE) {
        const int rr = i / BN, c = i % BN;
        const int r = m0 + p * MOE_MT + rr;
        if (r >= m_end) continue;
        float v = red[0][rr][c];
#pragma unroll
        for (int wv = 1; wv < MGWAVES; ++wv) v += red[wv][rr][c];
        if (direct) {
          out[(int64_t)r * N + n0 + c] = f2bf(scale ? v * scale[r] : v);
        } else {
          work[(int64_t)r * N + n0 + c] = v;
        }
      }
    }
  }
}

// gate/up slab -> silu(gate)*up bf16.  work rows are [gate | up] pairs of
// F columns each (same layout as skinny_gemm's reduce_slabs_silu, splitk=1;
// duplicated here because -fno-gpu-rdc keeps __global__ symbols per-TU).
__global__ void moe_silu_kernel(__hip_bfloat16* __restrict__ out,  // [R,F]
                                const float* __restrict__ work,    // [R,2F]
                                const int R, const int F) {
  const int64_t total = (int64_t)R * F;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int r = i / F, c = i % F;
    const float* row = work + (int64_t)r * 2 * F;
    const float g = row[c], u = row[F + c];
    const float s = g / (1.f + __expf(-g));
    out[i] = f2bf(s * u);
  }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cstdlib>

static int moe_env_int(const char* name, int dflt) {
  const char* v = getenv(name);
  return v ? atoi(v) : dflt;
}

static void launch_moe(torch::Tensor work, torch::Tensor out, torch::Tensor x,
                       torch::Tensor w, const int* tok, torch::Tensor off,
                       const float* scale, int direct, int R, hipStream_t s) {
  const int E = w.size(0), N = w.size(1), K = w.size(2);
  TORCH_CHECK(x.size(1) == K, "moe_gemm: K mismatch");
  TORCH_CHECK(N % BN == 0, "moe_gemm: N must be a multiple of 64");
  TORCH_CHECK(off.numel() == E + 1, "moe_gemm: off must be [E+1]");
  TORCH_CHECK(off.scalar_type() == torch::kInt32);
  const int wave_k = K / MGWAVES;
  dim3 grid(N / BN, E), block(MGWAVES * WAVE);
  // big-bucket regime: avg rows/expert >= ~40 makes >64-row buckets (and
  // their weight re-streams) common, so accumulate 128 rows per pass
  int mt = moe_env_int("CALF_MOE_MT", R >= moe_env_int("CALF_MOE_MT_AT", 320)
                                          ? 128 : 64);
  // MT=128/KU=64 sits at 172 VGPRs -> 1 wave/SIMD and LOSES across the
  // board; KU=32 keeps 2 waves/SIMD and wins the big-bucket regime
  // (profiles/r01_moe_micro.md sweep: T=256 1.24 -> 1.02 ms)
  int ku = moe_env_int("CALF_MOE_KU", mt == 128 ? 32 : 64);
  while (ku > 32 && wave_k % ku != 0) ku /= 2;
#define L(KU, MT)                                                       \
  hipLaunchKernelGGL((moe_gemm_kernel<KU, MT>), grid, block, 0, s,      \
                     (float*)work.data_ptr(),                           \
                     (__hip_bfloat16*)out.data_ptr(),                   \
                     (const __hip_bfloat16*)x.data_ptr(),               \
                     (const __hip_bfloat16*)w.data_ptr(), tok,          \
                     (const int*)off.data_ptr(), scale, N, K, direct)
  TORCH_CHECK(wave_k % 32 == 0, "moe_gemm: K=", K, " not divisible by 128");
  if (mt == 128) {
    if (ku >= 64) L(64, 128); else L(32, 128);
  } else {
    if (ku >= 64) L(64, 64); else L(32, 64);
  }
#undef L
  HIP_CHECK_LAST();
}

void moe_gateup_silu(torch::Tensor act, torch::Tensor x, torch::Tensor w,
                     torch::Tensor tok, torch::Tensor off,
                     torch::Tensor work) {
  // act[R,F] = silu(g)*u where [g|u] rows = x[tok[r]] @ w[expert(r)]^T,
  // w is [E, 2F, K] stacked gate|up, work is fp32 scratch >= R*2F.
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && act.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(tok.scalar_type() == torch::kInt32);
  const int R = tok.numel();
  const int N = w.size(1), F = N / 2;
  TORCH_CHECK(act.size(0) == R && act.size(1) == F);
  TORCH_CHECK(work.numel() >= (int64_t)R * N, "moe workspace too small");
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  launch_moe(work, act, x, w, (const int*)tok.data_ptr(), off, nullptr,
             /*direct=*/0, R, s);
  const int64_t total = (int64_t)R * F;
  const int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(moe_silu_kernel, dim3(blocks), dim3(256), 0, s,
                     (__hip_bfloat16*)act.data_ptr(),
                     (const float*)work.data_ptr(), R, F);
  HIP_CHECK_LAST();
}

void moe_down(torch::Tensor y, torch::Tensor act, torch::Tensor w,
              torch::Tensor off, torch::Tensor scale) {
  // y[R,d] = scale[r] * (act[r] @ w[expert(r)]^T); act rows already bucket
  // ordered (identity gather), w is [E, d, F], scale fp32 [R].
  TORCH_CHECK(act.is_contiguous() && w.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(act.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(scale.scalar_type() == torch::kFloat32);
  const int R = act.size(0);
  TORCH_CHECK(y.size(0) == R && y.size(1) == w.size(1));
  TORCH_CHECK(scale.numel() == R);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  launch_moe(y /*unused in direct mode: any fp32-typed arg*/, y, act, w,
             nullptr, off, (const float*)scale.data_ptr(), /*direct=*/1, R, s);
}

// ---------------------------------------------------------------------------
// Fused router: softmax -> top-K -> renormalize -> stable counting sort by
// expert, in ONE launch.  Replaces the torch pipeline (softmax, topk,
// argsort, scatter_add, cumsum, two gathers — ~8-10 kernels, ~50 us/layer
// of the MoE decode step in profiles/r02_moe_kernelstats.md) with a ~5 us
// kernel.  Deterministic by construction: per-expert placement scans flat
// entries in ascending index order, exactly the stable argsort the torch
// path produced, so downstream index_add sees the same row order.
// One block; T <= 256 (the grouped decode gate), E <= 16, K <= 4.
__global__ __launch_bounds__(256, 1) void moe_route_kernel(
    int32_t* __restrict__ tok,     // [T*K] source token per bucket row
    int32_t* __restrict__ off,     // [E+1] bucket offsets
    float* __restrict__ scale,     // [T*K] renormalized router weight
    int32_t* __restrict__ inv,     // [T*K] bucket row of flat entry t*K+j
    const float* __restrict__ logits,  // [T, E] fp32
    const int T, const int E, const int K) {
  __shared__ int16_t eids[256 * 4];
  __shared__ float ws[256 * 4];
  __shared__ int counts[17];
  __shared__ int offs[17];
  const int t = threadIdx.x;

  if (t < T) {
    float v[16];
    float mx = -INFINITY;
    for (int e = 0; e < E; ++e) {
      v[e] = logits[t * E + e];
      mx = fmaxf(mx, v[e]);
    }
    float denom = 0.f;
    for (int e = 0; e < E; ++e) {
      v[e] = __expf(v[e] - mx);
      denom += v[e];
    }
    // top-K selection (descending, matching torch.topk's order) + renorm
    float wsum = 0.f;
    for (int j = 0; j < K; ++j) {
      int best = -1;
      float bw = -1.f;
      for (int e = 0; e < E; ++e) {
        if (v[e] > bw) {
          bw = v[e];
          best = e;
        }
      }
      eids[t * K + j] = (int16_t)best;
      ws[t * K + j] = bw / denom;
      wsum += bw / denom;
      v[best] = -1.f;  // remove from later picks
    }
    for (int j = 0; j < K; ++j) ws[t * K + j] /= wsum;
  }
  if (t <= E) counts[t] = 0;
  __syncthreads();

  const int R = T * K;
  if (t < E) {  // per-expert count (serial: R <= 1024, E-way parallel)
    int c = 0;
    for (int i = 0; i < R; ++i) c += (eids[i] == t);
    counts[t] = c;
  }
  __syncthreads();
  if (t == 0) {
    int acc = 0;
    for (int e = 0; e <= E; ++e) {
      offs[e] = acc;
      off[e] = acc;
      if (e < E) acc += counts[e];
    }
  }
  __syncthreads();
  if (t < E) {  // stable placement in ascending flat-index order
    int pos = offs[t];
    for (int i = 0; i < R; ++i) {
      if (eids[i] == t) {
        tok[pos] = i / K;
        scale[pos] = ws[i];
        inv[i] = pos;
        ++pos;
      }
    }
  }
}

// out[t] = sum_j y[inv[t*K+j]] in fp32 — the deterministic replacement
// for zeros + index_add_ + two dtype converts (4 launches -> 1).
__global__ void moe_combine_kernel(__hip_bfloat16* __restrict__ out,  // [T,d]
                                   const __hip_bfloat16* __restrict__ y,  // [R,d]
                                   const int32_t* __restrict__ inv,   // [T*K]
                                   const int d, const int K) {
  const int t = blockIdx.x;
  int rows[4];
  for (int j = 0; j < K; ++j) rows[j] = inv[t * K + j];
  for (int c = threadIdx.x; c < d; c += blockDim.x) {
    float acc = 0.f;
    for (int j = 0; j < K; ++j)
      acc += bf2f(y[(int64_t)rows[j] * d + c]);
    out[(int64_t)t * d + c] = f2bf(acc);
  }
}

void moe_route(torch::Tensor tok, torch::Tensor off, torch::Tensor scale,
               torch::Tensor inv, torch::Tensor logits, int64_t topk) {
  TORCH_CHECK(logits.is_contiguous() && logits.scalar_type() == torch::kFloat32);
  const int T = logits.size(0), E = logits.size(1);
  TORCH_CHECK(T <= 256 && E <= 16 && topk <= 4, "moe_route: T<=256, E<=16, K<=4");
  TORCH_CHECK(tok.numel() >= T * topk && scale.numel() >= T * topk);
  TORCH_CHECK(off.numel() >= E + 1);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(moe_route_kernel, dim3(1), dim3(256), 0, s,
                     (int32_t*)tok.data_ptr(), (int32_t*)off.data_ptr(),
                     (float*)scale.data_ptr(), (int32_t*)inv.data_ptr(),
                     (const float*)logits.data_ptr(), T, E, (int)topk);
  HIP_CHECK_LAST();
}

void moe_combine(torch::Tensor out, torch::Tensor y, torch::Tensor inv,
                 int64_t topk) {
  TORCH_CHECK(out.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16 &&
              y.scalar_type() == torch::kBFloat16);
  const int T = out.size(0), d = out.size(1);
  TORCH_CHECK(inv.numel() >= T * topk && topk <= 4);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(moe_combine_kernel, dim3(T), dim3(256), 0, s,
                     (__hip_bfloat16*)out.data_ptr(),
                     (const __hip_bfloat16*)y.data_ptr(),
                     (const int32_t*)inv.data_ptr(), d, (int)topk);
  HIP_CHECK_LAST();
}
