// Tiled MFMA GEMM for the serving decode batches (M = 65..512+) on MI355X.
//
// out[M,N] = x[M,K] @ W^T, W stored [N,K] (nn.Linear layout), bf16 in,
// bf16 out (direct) or fp32 split-K slabs (work[sk][Mp][N], the same slab
// contract the skinny path's fused consumers read — rope_qkv_write_slab /
// fused_add_rmsnorm_slab in elementwise.hip).
//
// Why this kernel exists: at the headline batch (512) the
// projections ran hipBLASLt at 0.65-0.84 PF/s — fill/latency-bound, far
// from both the MFMA roof and the HBM roof (profiles/gemm_algo_table.json:
// qkv 32.8 us, o 26.5, down 72).  The shapes have only 32-96 natural
// 256-wide tiles on a 256-CU chip, so the lever is per-block LATENCY via
// split-K (guide: "(N/64)*SPLITK ~ 1/2-1x the CU count") plus a deep
// glds pipeline for per-CU throughput.
//
// Structure (CDNA4 guide §5 "256^2 8-phase template", adapted):
// - 256xBN tile (BN 128/256), BK=64, 8 waves (512 thr), 2 LDS buffers.
// - Both operands stage global->LDS by __builtin_amdgcn_global_load_lds
//   (16 B/lane), LDS image lane-linear, XOR swizzle applied to the SOURCE
//   address and re-applied on the ds_read_b128 side (guide §5.4 rule 21,
//   T2: byte ^= (row&7)<<4 -> conflict-free column-slice reads).
// - 4 compute phases per K-tile (one per m-quadrant, 16 MFMAs each for
//   BN=256) separated by raw s_barriers with s_setprio(1) around the MFMA
//   cluster (T5); next tile's A/B stages issue in phases 0/1 so their HBM
//   latency hides under ~2 phases of MFMA before the boundary drain
//   (T3/T4 in spirit: loads in flight across the inner barriers; the
//   boundary __syncthreads() emits the vmcnt(0) that publishes the LDS
//   writes - the only full drain per K-tile).
// - XCD-bijective block remap (T1) keeps one n-panel's blocks on one XCD
//   so the W panel is read from that XCD's L2 by its m/sk siblings.
//
// Measured on gfx950 (profiles/r02_tile_gemm.md): see routing notes in
// models/llama.py::_tile_plan.

#include "common.h"

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;

#define MFMA16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

template <int BN>
__global__ __launch_bounds__(512, 2) void tile_gemm_kernel(
    float* __restrict__ work,             // [splitk, Mp, N] fp32 slabs
    __hip_bfloat16* __restrict__ out,     // [M, N] (direct == 1)
    const __hip_bfloat16* __restrict__ x, // [M, K]
    const __hip_bfloat16* __restrict__ w, // [N, K]
    const int M, const int N, const int K, const int m_tiles,
    const int splitk, const int direct, const int Mp) {
  constexpr int BM = 256, BK = 64;
  constexpr int WC = (BN == 256) ? 4 : 2;  // waves along N
  constexpr int MI = BM / 16 / (8 / WC);   // m-frags per wave (8 / 4)
  constexpr int NI = BN / 16 / WC;         // n-frags per wave (4)
  constexpr int MQ = MI / 4;               // m-frags per phase
  constexpr int A_ELEMS = BM * BK;
  constexpr int B_ELEMS = BN * BK;
  constexpr int BUF = A_ELEMS + B_ELEMS;
  __shared__ __attribute__((aligned(16))) __hip_bfloat16 smem[2 * BUF];

  // ---- block id -> (n_tile, m_tile, sk) with XCD-bijective remap (T1):
  // consecutive remapped ids share the n panel, and land on one XCD
  int bid = blockIdx.x;
  {
    const int nwg = gridDim.x, q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, i = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  const int per_n = m_tiles * splitk;
  const int n_t = bid / per_n, rem = bid % per_n;
  const int m_t = rem / splitk, sk = rem % splitk;
  const int n0 = n_t * BN, m0 = m_t * BM;
  const int kb = K / splitk;
  const int k0 = sk * kb;
  const int nt = kb / BK;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE, lane = tid % WAVE;
  const int l16 = lane & 15, l4 = lane >> 4;
  // staging: thread t covers image row sr (8 rows per wave per call),
  // 16B chunk sc of the 128 B row; source chunk is XOR-swizzled (rule 21)
  const int sr = tid / 8;  // == wid*8 + lane/8
  const int sc = tid % 8;

  const auto stage_a = [&](int buf, int ktile) {
    const int kk = k0 + ktile * BK;
#pragma unroll
    for (int c = 0; c < BM / 64; ++c) {
      const int rloc = c * 64 + sr;
      const int grow = min(m0 + rloc, M - 1);  // tail tile row clamp
      const int chunk = sc ^ (rloc & 7);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(
              x + (int64_t)grow * K + kk + chunk * 8),
          reinterpret_cast<unsigned int*>(
              &smem[buf * BUF + (c * 64 + wid * 8) * BK]),
          16, 0, 0);
    }
  };
  const auto stage_b = [&](int buf, int ktile) {
    const int kk = k0 + ktile * BK;
#pragma unroll
    for (int c = 0; c < BN / 64; ++c) {
      const int rloc = c * 64 + sr;
      const int chunk = sc ^ (rloc & 7);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(
              w + (int64_t)(n0 + rloc) * K + kk + chunk * 8),
          reinterpret_cast<unsigned int*>(
              &smem[buf * BUF + A_ELEMS + (c * 64 + wid * 8) * BK]),
          16, 0, 0);
    }
  };
  // fragment reads: row varies over the 16-lane group at one k-chunk ->
  // the T2 column-slice pattern; same XOR as the staged source
  const auto read_a = [&](int buf, int mi, int ks) {
    const int row = (wid / WC) * (16 * MI) + mi * 16 + l16;
    const int chunk = (ks * 4 + l4) ^ (row & 7);
    return *reinterpret_cast<const bf16x8v*>(
        &smem[buf * BUF + row * BK + chunk * 8]);
  };
  const auto read_b = [&](int buf, int ni, int ks) {
    const int row = (wid % WC) * (16 * NI) + ni * 16 + l16;
    const int chunk = (ks * 4 + l4) ^ (row & 7);
    return *reinterpret_cast<const bf16x8v*>(
        &smem[buf * BUF + A_ELEMS + row * BK + chunk * 8]);
  };

  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = (f32x4){0.f, 0.f, 0.f, 0.f};

  stage_a(0, 0);
  stage_b(0, 0);
  __syncthreads();  // drains the glds (vmcnt 0) and publishes LDS

  int cur = 0;
  for (int t = 0; t < nt; ++t) {
    bf16x8v bfr[NI][2];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      // issue next tile's stages early (phases 0/1): ~2 phases of MFMA
      // cover the HBM latency before the boundary drain below
      if (q == 0 && t + 1 < nt) stage_a(cur ^ 1, t + 1);
      if (q == 1 && t + 1 < nt) stage_b(cur ^ 1, t + 1);
      if (q == 0) {
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) bfr[ni][ks] = read_b(cur, ni, ks);
      }
      bf16x8v afr[MQ][2];
#pragma unroll
      for (int mi = 0; mi < MQ; ++mi)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          afr[mi][ks] = read_a(cur, q * MQ + mi, ks);
      __builtin_amdgcn_s_barrier();  // raw: do NOT drain the in-flight glds
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int mi = 0; mi < MQ; ++mi)
#pragma unroll
          for (int ni = 0; ni < NI; ++ni)
            acc[q * MQ + mi][ni] =
                MFMA16(afr[mi][ks], bfr[ni][ks], acc[q * MQ + mi][ni]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    __syncthreads();  // K-tile boundary: vmcnt(0) publishes next tile's LDS
    cur ^= 1;
  }

  // ---- epilogue.  C layout (mfma_f32_16x16x32): col = l16, row = l4*4+r.
  const int r0 = (wid / WC) * (16 * MI);
  const int c0 = (wid % WC) * (16 * NI);
  if (direct) {
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = m0 + r0 + mi * 16 + l4 * 4 + r;
          if (row < M)
            out[(int64_t)row * N + n0 + c0 + ni * 16 + l16] =
                f2bf(acc[mi][ni][r]);
        }
  } else {
    float* slab = work + (int64_t)sk * Mp * N;
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = m0 + r0 + mi * 16 + l4 * 4 + r;
          slab[(int64_t)row * N + n0 + c0 + ni * 16 + l16] = acc[mi][ni][r];
        }
  }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// fwd decl (skinny_gemm.hip)
__global__ void reduce_slabs_kernel(__hip_bfloat16* __restrict__ out,
                                    const float* __restrict__ work,
                                    const int M, const int N, const int Mp,
                                    const int splitk);

static void launch_tile(torch::Tensor work, torch::Tensor out,
                        torch::Tensor x, torch::Tensor w, int64_t bn,
                        int64_t splitk, int direct, hipStream_t s) {
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  const int m_tiles = (M + 255) / 256;
  const int Mp = m_tiles * 256;
  TORCH_CHECK(N % bn == 0, "tile_gemm: N % BN != 0");
  TORCH_CHECK((K / splitk) % 64 == 0 && K % splitk == 0,
              "tile_gemm: K must split into 64-deep tiles");
  const int blocks = (N / (int)bn) * m_tiles * (int)splitk;
  TORCH_CHECK(direct || work.numel() >= (int64_t)splitk * Mp * N,
              "tile_gemm: workspace too small");
#define LT(BNV)                                                              \
  hipLaunchKernelGGL((tile_gemm_kernel<BNV>), dim3(blocks), dim3(512), 0, s, \
                     (float*)work.data_ptr(),                                \
                     (__hip_bfloat16*)out.data_ptr(),                        \
                     (const __hip_bfloat16*)x.data_ptr(),                    \
                     (const __hip_bfloat16*)w.data_ptr(), M, N, K, m_tiles,  \
                     (int)splitk, direct, Mp)
  if (bn == 256)
    LT(256);
  else
    LT(128);
#undef LT
}

void tile_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
               torch::Tensor work, int64_t bn, int64_t splitk) {
  // out[M,N] = x[M,K] @ w[N,K]^T; splitk>1 reduces fp32 slabs -> bf16
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int M = x.size(0), N = w.size(0);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  launch_tile(work, out, x, w, bn, splitk, splitk == 1 ? 1 : 0, s);
  if (splitk > 1) {
    const int Mp = ((M + 255) / 256) * 256;
    const int64_t total = (int64_t)M * N;
    const int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
    hipLaunchKernelGGL(reduce_slabs_kernel, dim3(blocks), dim3(256), 0, s,
                       (__hip_bfloat16*)out.data_ptr(),
                       (const float*)work.data_ptr(), M, N, Mp, (int)splitk);
  }
  HIP_CHECK_LAST();
}

void tile_gemm_slabs(torch::Tensor x, torch::Tensor w, torch::Tensor work,
                     int64_t bn, int64_t splitk) {
  // slab-only launch: a fused consumer (rope_qkv_write_slab /
  // fused_add_rmsnorm_slab) folds the split-K reduce into its own pass
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  launch_tile(work, x /*unused*/, x, w, bn, splitk, /*direct=*/0, s);
  HIP_CHECK_LAST();
}
