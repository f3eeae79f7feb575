// Causal GQA prefill attention (flash-style) on MFMA for MI355X (gfx950).
//
// Decomposition: grid = (Hq, ceil(S/64)); workgroup = 4 waves, 64 q-rows per
// workgroup, 16 q-rows per wave.  K/V tiles of 32 keys stream through LDS;
// online softmax per q-row; P·V accumulates in MFMA accumulators.
//
// MFMA: v_mfma_f32_16x16x32_bf16.  Fragment maps (verified by the gpu-lane
// identity/asymmetric tests in tests/unit/test_ops_gpu.py):
//   A[i][k]: lane holds i = lane&15, k = (lane>>4)*8 + m  (m = 0..7)
//   B[k][j]: lane holds j = lane&15, k = (lane>>4)*8 + m
//   C/D[r][c]: lane holds c = lane&15, r = (lane>>4)*4 + reg
//
// LDS layout choices (Guideline 4):
//   K tile  [32][128] bf16, XOR-swizzled byte ^= (row&15)<<4 — the B-operand
//           read is 16 lanes reading 16 different rows at one col-range, the
//           16-way-conflict pattern; the swizzle makes it conflict-free.
//   V tile  stored TRANSPOSED [128][32+8pad] bf16 (80 B rows) so the PV
//           B-fragment (8 consecutive KEYS at one dim) is a contiguous
//           ds_read_b128; the 8-halfword pad makes the 16-lane group hit 16
//           distinct banks (stride 20 words, gcd-free mod 64).
//   P tile  per-wave [16][32] bf16 round-trip: the QK C-layout is not the PV
//           A-layout, so P crosses LDS once per tile.
//
// Q is consumed straight from global memory (A-fragments are contiguous
// 16 B per lane) — no LDS staging for the operand that is read once.

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
    __hip_bfloat16* __restrict__ out,       // [S, Hq, HD]
    const __hip_bfloat16* __restrict__ q,   // [S, Hq, HD]
    const __hip_bfloat16* __restrict__ k,   // [S, Hkv, HD]
    const __hip_bfloat16* __restrict__ v,   // [S, Hkv, HD]
    const int S, const int hq, const int hkv, const float scale,
    const int past_len) {                   // keys 0..past_len-1 are context
  const int head = blockIdx.x;
  const int kvh = head / (hq / hkv);
  const int qb = blockIdx.y;                 // q-row block of 64
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int l16 = lane & 15;                 // fragment row/col index
  const int l4 = lane >> 4;                  // fragment k-group

  __shared__ __hip_bfloat16 k_lds[KVBLK * HD];          // swizzled
  __shared__ __hip_bfloat16 vt_lds[HD * VT_PITCH];      // transposed + padded
  __shared__ __hip_bfloat16 p_lds[NW][16 * KVBLK];

  // ---- load this wave's Q fragments once: rows qb*64 + wid*16 + (0..15)
  const int qrow0 = qb * QBLK + wid * 16;
  bf16x8v qfrag[4];  // 4 k-steps of 32 over HD=128
  {
    const int row = qrow0 + l16;
    const __hip_bfloat16* src =
        q + ((int64_t)min(row, S - 1) * hq + head) * HD + l4 * 8;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks)
      qfrag[ks] = *reinterpret_cast<const bf16x8v*>(src + ks * 32);
  }

  // ---- per-row softmax state: each lane carries 4 rows (C-layout rows)
  float mrow[4], lrow[4];
  f32x4 o_acc[8];  // O[16][128]: 8 dim-tiles of 16x16
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    mrow[r] = -INFINITY;
    lrow[r] = 0.f;
  }
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) o_acc[dt] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // causal bound: this block's highest q sees keys < past_len + qb*64 + 64
  const int kv_limit = min(S + past_len, past_len + (qb + 1) * QBLK);
  const int ntiles = (kv_limit + KVBLK - 1) / KVBLK;

  for (int tile = 0; tile < ntiles; ++tile) {
    const int key0 = tile * KVBLK;
    // ---- stage K tile (swizzled rows) and V^T tile cooperatively
    // 256 threads, K tile has 32 rows x 16 bf16x8 chunks = 512 chunks
    for (int c = threadIdx.x; c < KVBLK * (HD / 8); c += NW * WAVE) {
      const int row = c / (HD / 8), d8 = c % (HD / 8);
      const int key = key0 + row;
      bf16x8 val;
      if (key < kv_limit) {
        val = *reinterpret_cast<const bf16x8*>(
            k + ((int64_t)key * hkv + kvh) * HD + d8 * 8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) val.v[j] = f2bf(0.f);
      }
      // swizzle: byte ^= (row&15)<<4  == halfword ^= (row&15)<<3
      const int hw = (row * HD + d8 * 8) ^ ((row & 15) << 3);
      *reinterpret_cast<bf16x8*>(&k_lds[hw]) = val;
      // V^T: scatter the 8 dims of this key
      bf16x8 vv;
      if (key < kv_limit) {
        vv = *reinterpret_cast<const bf16x8*>(
            v + ((int64_t)key * hkv + kvh) * HD + d8 * 8);
      } else {
        vv = val;  // zeros
      }
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
        // B-frag: key j = jt*16 + l16, dims k = ks*32 + l4*8 .. +8
        const int row = jt * 16 + l16;
        const int hw = (row * HD + ks * 32 + l4 * 8) ^ ((row & 15) << 3);
        bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(&k_lds[hw]);
        sacc[jt] = MFMA16(qfrag[ks], bfrag, sacc[jt]);
      }
    }

    // ---- online softmax on the 16x32 score block (C-layout)
    // lane's 4 rows r: global q row = qrow0 + (l4*4 + reg)
    float p[2][4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int qrow = qrow0 + l4 * 4 + reg;
      float s0 = sacc[0][reg] * scale;
      float s1 = sacc[1][reg] * scale;
      // causal mask: key index key0 + jt*16 + l16 must be <= past_len + qrow
      if (key0 + l16 > past_len + qrow || qrow >= S) s0 = -INFINITY;
      if (key0 + 16 + l16 > past_len + qrow || qrow >= S) s1 = -INFINITY;
      // row max over the 16 lanes of this quarter (both j-tiles)
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
      // rescale O accumulators for this row (row lives at acc index reg)
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) o_acc[dt][reg] *= rsc;
    }

    // ---- P round-trip: C-layout -> LDS -> A-layout fragments
    // write: lane has P[r = l4*4+reg][c = jt*16 + l16]
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
#pragma unroll
      for (int jt = 0; jt < 2; ++jt)
        p_lds[wid][(l4 * 4 + reg) * KVBLK + jt * 16 + l16] = f2bf(p[jt][reg]);
    }
    __syncthreads();  // also covers the K/V tile reuse below

    // read A-frag: P[i = l16][k = l4*8 + m], contiguous 16 B
    bf16x8v pfrag = *reinterpret_cast<const bf16x8v*>(
        &p_lds[wid][l16 * KVBLK + l4 * 8]);

    // ---- O += P·V : 8 dim-tiles, one MFMA each (K = 32 keys)
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      // B-frag: V[k = l4*8 + m][j = dt*16 + l16] from V^T rows (dim-major)
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
    __hip_bfloat16* orow = out + ((int64_t)qrow * hq + head) * HD;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt)
      orow[dt * 16 + l16] = f2bf(o_acc[dt][reg] * inv);
  }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                  torch::Tensor v, double scale, int64_t past_len) {
  // q/out: [S, Hq, HD]; k/v: [past_len+S, Hkv, HD] (context + new keys)
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(out.is_contiguous());
  const int S = q.size(0), hq = q.size(1), dh = q.size(2);
  const int hkv = k.size(1);
  TORCH_CHECK(dh == HD, "prefill kernel assumes head_dim == 128");
  TORCH_CHECK(k.size(0) == past_len + S, "k must hold past + new keys");
  dim3 grid(hq, (S + QBLK - 1) / QBLK), block(NW * WAVE);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(attn_prefill_kernel, grid, block, 0, s,
                     (__hip_bfloat16*)out.data_ptr(),
                     (const __hip_bfloat16*)q.data_ptr(),
                     (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(), S, hq, hkv,
                     (float)scale, (int)past_len);
  HIP_CHECK_LAST();
}
