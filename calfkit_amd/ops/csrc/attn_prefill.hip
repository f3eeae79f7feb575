// Causal GQA prefill attention (flash-style) on MFMA for MI355X (gfx950),
// varlen-packed: many sequences prefill in ONE launch (the agent engine
// packs every waiting prompt into one forward).
//
// Decomposition: grid = (Hq, n_qblocks_total); workgroup = 4 waves, 64
// q-rows per workgroup, 16 q-rows per wave.  block_map names (seq, qblock)
// per workgroup; per-seq row ranges come from q_starts/kv_starts prefix
// arrays.  K/V tiles of 32 keys stream through LDS; online softmax per
// q-row; P·V accumulates in MFMA accumulators.
//
// MFMA: v_mfma_f32_16x16x32_bf16.  Fragment maps (verified by the gpu-lane
// tests in tests/unit/test_ops_gpu.py against fp32 torch references):
//   A[i][k]: lane holds i = lane&15, k = (lane>>4)*8 + m  (m = 0..7)
//   B[k][j]: lane holds j = lane&15, k = (lane>>4)*8 + m
//   C/D[r][c]: lane holds c = lane&15, r = (lane>>4)*4 + reg
//
// LDS layout (Guideline 4):
//   K tile  [32][128] bf16, XOR-swizzled halfword ^= (row&15)<<3 — the
//           B-operand read is 16 lanes reading 16 different rows at one
//           col-range (the 16-way-conflict pattern); the swizzle zeroes it.
//   V tile  stored TRANSPOSED [128][32+8pad] bf16 (80 B rows) so the PV
//           B-fragment (8 consecutive KEYS at one dim) is one contiguous
//           ds_read_b128; the pad puts the 16-lane group on 16 distinct
//           banks (stride 20 words, coprime with 64).
//   P tile  per-wave [16][32] bf16 round-trip (QK C-layout != PV A-layout).

#include "common.h"

#define HD 128
#define QBLK 64
#define KVBLK 32
#define NW 4   // waves per workgroup

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;

#define MFMA16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

// V^T row pitch in halfwords: 32 keys + 8 pad = 40 (80 B)
#define VT_PITCH 40

__global__ __launch_bounds__(NW * WAVE) void attn_prefill_kernel(
    __hip_bfloat16* __restrict__ out,       // [Tq, Hq, HD] packed
    const __hip_bfloat16* __restrict__ q,   // [Tq, Hq, HD] packed
    const __hip_bfloat16* __restrict__ k,   // [Tkv, Hkv, HD] packed
    const __hip_bfloat16* __restrict__ v,   // [Tkv, Hkv, HD] packed
    const int32_t* __restrict__ block_map,  // [nblocks, 2] = (seq, qblock)
    const int32_t* __restrict__ q_starts,   // [B+1] prefix into q rows
    const int32_t* __restrict__ kv_starts,  // [B+1] prefix into kv rows
    const int32_t* __restrict__ past_lens,  // [B] context rows per seq
    const int hq, const int hkv, const float scale,
    const int q_stride, const int kv_stride) {  // row strides in elements
  // XCD-grouped head order (guide T1): block b lands on XCD b%8 and
  // blockIdx.x varies fastest, so the plain head=x order puts the G
  // q-heads of one kv-head on G DIFFERENT XCDs — each re-streams the
  // same K/V panel from HBM.  This bijection makes the heads of kv-head
  // g share x%hkv == g, i.e. one XCD (hkv == 8 on llama): 3 of the 4
  // panel reads become that XCD's L2 hits.
  const int head = (blockIdx.x % hkv) * (hq / hkv) + blockIdx.x / hkv;
  const int kvh = head / (hq / hkv);
  const int seq = block_map[blockIdx.y * 2];
  const int qb = block_map[blockIdx.y * 2 + 1];
  const int q0 = q_starts[seq];
  const int S = q_starts[seq + 1] - q0;       // new tokens in this seq
  const int kv0 = kv_starts[seq];
  const int past = past_lens[seq];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int l16 = lane & 15;                 // fragment row/col index
  const int l4 = lane >> 4;                  // fragment k-group

  __shared__ __hip_bfloat16 k_lds[KVBLK * HD];          // swizzled
  __shared__ __hip_bfloat16 vt_lds[HD * VT_PITCH];      // transposed + padded
  __shared__ __hip_bfloat16 p_lds[NW][16 * KVBLK];

  // ---- load this wave's Q fragments once: local rows qb*64 + wid*16 + 0..15
  const int qrow0 = qb * QBLK + wid * 16;    // local to the sequence
  bf16x8v qfrag[4];
  {
    const int row = min(qrow0 + l16, S - 1);
    const __hip_bfloat16* src =
        q + (int64_t)(q0 + row) * q_stride + head * HD + l4 * 8;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks)
      qfrag[ks] = *reinterpret_cast<const bf16x8v*>(src + ks * 32);
  }

  float mrow[4], lrow[4];
  f32x4 o_acc[8];  // O[16][128]: 8 dim-tiles of 16x16
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    mrow[r] = -INFINITY;
    lrow[r] = 0.f;
  }
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) o_acc[dt] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // causal bound: this block's highest q row sees keys < past + qb*64 + 64
  const int kv_limit = min(S + past, past + (qb + 1) * QBLK);
  const int ntiles = (kv_limit + KVBLK - 1) / KVBLK;

  for (int tile = 0; tile < ntiles; ++tile) {
    const int key0 = tile * KVBLK;
    // ---- stage K tile (swizzled rows) + V^T tile cooperatively
    for (int c = threadIdx.x; c < KVBLK * (HD / 8); c += NW * WAVE) {
      const int row = c / (HD / 8), d8 = c % (HD / 8);
      const int key = key0 + row;
      bf16x8 val, vv;
      if (key < kv_limit) {
        const int64_t kbase =
            (int64_t)(kv0 + key) * kv_stride + kvh * HD + d8 * 8;
        val = *reinterpret_cast<const bf16x8*>(k + kbase);
        vv = *reinterpret_cast<const bf16x8*>(v + kbase);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) val.v[j] = f2bf(0.f);
        vv = val;
      }
      const int hw = (row * HD + d8 * 8) ^ ((row & 15) << 3);
      *reinterpret_cast<bf16x8*>(&k_lds[hw]) = val;
#pragma unroll
      for (int j = 0; j < 8; ++j) vt_lds[(d8 * 8 + j) * VT_PITCH + row] = vv.v[j];
    }
    __syncthreads();

    // ---- S = Q·K^T for this wave's 16 rows x 32 keys (2 j-tiles)
    f32x4 sacc[2];
    sacc[0] = (f32x4){0.f, 0.f, 0.f, 0.f};
    sacc[1] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int jt = 0; jt < 2; ++jt) {
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const int row = jt * 16 + l16;
        const int hw = (row * HD + ks * 32 + l4 * 8) ^ ((row & 15) << 3);
        bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(&k_lds[hw]);
        sacc[jt] = MFMA16(qfrag[ks], bfrag, sacc[jt]);
      }
    }

    // ---- online softmax on the 16x32 score block (C-layout)
    float p[2][4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int qrow = qrow0 + l4 * 4 + reg;  // local q row
      float s0 = sacc[0][reg] * scale;
      float s1 = sacc[1][reg] * scale;
      if (key0 + l16 > past + qrow || qrow >= S) s0 = -INFINITY;
      if (key0 + 16 + l16 > past + qrow || qrow >= S) s1 = -INFINITY;
      float mt = fmaxf(s0, s1);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mt = fmaxf(mt, __shfl_xor(mt, off, WAVE));
      const float mn = fmaxf(mrow[reg], mt);
      const float rsc = (mrow[reg] == -INFINITY) ? 0.f : __expf(mrow[reg] - mn);
      const float p0 = (s0 == -INFINITY) ? 0.f : __expf(s0 - mn);
      const float p1 = (s1 == -INFINITY) ? 0.f : __expf(s1 - mn);
      float psum = p0 + p1;
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) psum += __shfl_xor(psum, off, WAVE);
      lrow[reg] = lrow[reg] * rsc + psum;
      mrow[reg] = mn;
      p[0][reg] = p0;
      p[1][reg] = p1;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) o_acc[dt][reg] *= rsc;
    }

    // ---- P round-trip: C-layout -> LDS -> A-layout fragments
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
#pragma unroll
      for (int jt = 0; jt < 2; ++jt)
        p_lds[wid][(l4 * 4 + reg) * KVBLK + jt * 16 + l16] = f2bf(p[jt][reg]);
    }
    __syncthreads();

    bf16x8v pfrag = *reinterpret_cast<const bf16x8v*>(
        &p_lds[wid][l16 * KVBLK + l4 * 8]);

    // ---- O += P·V : 8 dim-tiles, one MFMA each (K = 32 keys)
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      bf16x8v vfrag = *reinterpret_cast<const bf16x8v*>(
          &vt_lds[(dt * 16 + l16) * VT_PITCH + l4 * 8]);
      o_acc[dt] = MFMA16(pfrag, vfrag, o_acc[dt]);
    }
    __syncthreads();  // tile buffers free for the next iteration
  }

  // ---- epilogue: normalize rows and store
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int qrow = qrow0 + l4 * 4 + reg;
    if (qrow >= S) continue;
    const float inv = (lrow[reg] > 0.f) ? 1.f / lrow[reg] : 0.f;
    __hip_bfloat16* orow = out + ((int64_t)(q0 + qrow) * hq + head) * HD;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt)
      orow[dt * 16 + l16] = f2bf(o_acc[dt][reg] * inv);
  }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void attn_prefill_varlen(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                         torch::Tensor v, torch::Tensor block_map,
                         torch::Tensor q_starts, torch::Tensor kv_starts,
                         torch::Tensor past_lens, double scale) {
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == k.size(2));
  TORCH_CHECK(k.stride(0) == v.stride(0) && k.stride(1) == v.stride(1));
  TORCH_CHECK(block_map.scalar_type() == torch::kInt32);
  TORCH_CHECK(q_starts.scalar_type() == torch::kInt32);
  const int hq = q.size(1), dh = q.size(2);
  const int hkv = k.size(1);
  const int nblocks = block_map.size(0);
  TORCH_CHECK(dh == HD, "prefill kernel assumes head_dim == 128");
  dim3 grid(hq, nblocks), block(NW * WAVE);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(attn_prefill_kernel, grid, block, 0, s,
                     (__hip_bfloat16*)out.data_ptr(),
                     (const __hip_bfloat16*)q.data_ptr(),
                     (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(),
                     (const int32_t*)block_map.data_ptr(),
                     (const int32_t*)q_starts.data_ptr(),
                     (const int32_t*)kv_starts.data_ptr(),
                     (const int32_t*)past_lens.data_ptr(), hq, hkv,
                     (float)scale, (int)q.stride(0), (int)k.stride(0));
  HIP_CHECK_LAST();
}

void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                  torch::Tensor v, double scale, int64_t past_len) {
  // single-sequence convenience wrapper over the varlen kernel
  const int S = q.size(0);
  const int nblocks = (S + QBLK - 1) / QBLK;
  auto opts = torch::TensorOptions().dtype(torch::kInt32).device(q.device());
  auto block_map = torch::empty({nblocks, 2}, opts);
  block_map.select(1, 0).zero_();
  block_map.select(1, 1).copy_(
      torch::arange(nblocks, torch::TensorOptions().dtype(torch::kInt32)));
  auto q_starts = torch::tensor({0, S}, opts);
  auto kv_starts = torch::tensor({0, (int)k.size(0)}, opts);
  auto past = torch::tensor({(int)past_len}, opts);
  TORCH_CHECK(k.size(0) == past_len + S, "k must hold past + new keys");
  attn_prefill_varlen(out, q, k, v, block_map, q_starts, kv_starts, past, scale);
}
