// Fused elementwise / normalization kernels for the Llama-family decode and
// prefill paths on MI355X (gfx950).  All memory-bound: bf16 I/O is vectorized
// as 16 B/lane (bf16x8) per Guideline 13; reductions are wave-wide shuffles.
//
// Kernels:
//   rmsnorm           out = x * rsqrt(mean(x^2)+eps) * w
//   fused_add_rmsnorm residual += hidden; hidden = rmsnorm(residual)
//   silu_mul          out = silu(gate) * up       (gate/up packed [T, 2F])
//   rope_qk           in-place neox rotate-half rotary on Q and K rows
//   kv_write          scatter new K/V rows into the paged cache
//
// Capability parity note: the reference (calf-ai/calfkit-sdk) has no kernels —
// these implement the LocalModelClient compute of BASELINE.json's north star.

#include "common.h"

// ------------------------------------------------------------------ rmsnorm

template <int BLOCK>
__global__ void rmsnorm_kernel(__hip_bfloat16* __restrict__ out,
                               const __hip_bfloat16* __restrict__ inp,
                               const __hip_bfloat16* __restrict__ weight,
                               const int d, const float eps) {
  __shared__ float red[BLOCK / WAVE];
  const int64_t row = blockIdx.x;
  const __hip_bfloat16* x = inp + row * d;
  __hip_bfloat16* o = out + row * d;

  float ss = 0.f;
  const int nvec = d / 8;
  const bf16x8* xv = reinterpret_cast<const bf16x8*>(x);
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    bf16x8 v = xv[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v.v[j]);
      ss += f * f;
    }
  }
  if constexpr (BLOCK == WAVE) {
    ss = wave_sum(ss);  // single-wave block: no LDS, no barrier
  } else {
    ss = block_sum<BLOCK>(ss, red);
  }
  const float inv = rsqrtf(ss / d + eps);

  const bf16x8* wv = reinterpret_cast<const bf16x8*>(weight);
  bf16x8* ov = reinterpret_cast<bf16x8*>(o);
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    bf16x8 v = xv[i], w = wv[i], r;
#pragma unroll
    for (int j = 0; j < 8; ++j) r.v[j] = f2bf(bf2f(v.v[j]) * inv * bf2f(w.v[j]));
    ov[i] = r;
  }
}

// residual += hidden; hidden = rmsnorm(residual) — one pass over HBM instead
// of three (the producing add is fused into the consuming norm).
// VPT: row vec8s cached in registers across the reduction, so the second
// pass never re-reads the row (the pre-r02 version re-read residual from
// L1 after the barrier and ran latency-bound at 9.1 us vs a ~2-3 us
// stream bound at batch 512).
template <int BLOCK>
__global__ void fused_add_rmsnorm_kernel(__hip_bfloat16* __restrict__ hidden,
                                         __hip_bfloat16* __restrict__ residual,
                                         const __hip_bfloat16* __restrict__ weight,
                                         const int d, const float eps) {
  // VPT scales so every BLOCK size covers d <= 8192 (70B hidden).
  // r02 PMC at BLOCK=256: 79.6% SQ_WAIT_ANY, 4.5% issue-stall — pure
  // HBM-latency parking; the A/B showed bigger blocks (fewer, wider
  // phases) beat more-independent smaller blocks.
  constexpr int VPT = (256 * 4) / BLOCK;
  __shared__ float red[BLOCK / WAVE];
  const int64_t row = blockIdx.x;
  __hip_bfloat16* h = hidden + row * d;
  __hip_bfloat16* r = residual + row * d;

  float ss = 0.f;
  const int nvec = d / 8;
  bf16x8* hv = reinterpret_cast<bf16x8*>(h);
  bf16x8* rv = reinterpret_cast<bf16x8*>(r);
  bf16x8 keep[VPT];
#pragma unroll
  for (int c = 0; c < VPT; ++c) {
    const int i = threadIdx.x + c * BLOCK;
    if (i >= nvec) break;
    bf16x8 a = hv[i], b = rv[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(a.v[j]) + bf2f(b.v[j]);
      b.v[j] = f2bf(f);
      ss += f * f;
    }
    keep[c] = b;
    rv[i] = b;  // residual holds the sum
  }
  if constexpr (BLOCK == WAVE) {
    ss = wave_sum(ss);  // single-wave block: no LDS, no barrier
  } else {
    ss = block_sum<BLOCK>(ss, red);
  }
  const float inv = rsqrtf(ss / d + eps);

  const bf16x8* wv = reinterpret_cast<const bf16x8*>(weight);
#pragma unroll
  for (int c = 0; c < VPT; ++c) {
    const int i = threadIdx.x + c * BLOCK;
    if (i >= nvec) break;
    bf16x8 b = keep[c], w = wv[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = f2bf(bf2f(b.v[j]) * inv * bf2f(w.v[j]));
    hv[i] = o;
  }
}

// ----------------------------------------------------------------- silu_mul

__global__ void silu_mul_kernel(__hip_bfloat16* __restrict__ out,
                                const __hip_bfloat16* __restrict__ gateup,
                                const int64_t rows, const int f) {
  // gateup row layout: [gate (f) | up (f)]
  const int64_t row = blockIdx.y;
  const __hip_bfloat16* g = gateup + row * (2 * (int64_t)f);
  const __hip_bfloat16* u = g + f;
  __hip_bfloat16* o = out + row * f;
  const int nvec = f / 8;
  const bf16x8* gv = reinterpret_cast<const bf16x8*>(g);
  const bf16x8* uv = reinterpret_cast<const bf16x8*>(u);
  bf16x8* ov = reinterpret_cast<bf16x8*>(o);
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += gridDim.x * blockDim.x) {
    bf16x8 a = gv[i], b = uv[i], r;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = bf2f(a.v[j]);
      // rcp instead of a full-precision divide (1-ulp rcp is plenty for a
      // bf16 result; the IEEE div expands to ~10 VALU ops per element)
      float s = x * __builtin_amdgcn_rcpf(1.f + __expf(-x));
      r.v[j] = f2bf(s * bf2f(b.v[j]));
    }
    ov[i] = r;
  }
}

// ------------------------------------------------------------------- rope

// Neox rotate-half RoPE, in place, on packed [T, H, dh] Q and K.
// cos_sin: [max_pos, dh] laid out as [cos(dh/2) | sin(dh/2)] per position
// (host-precomputed f32 table — on-device trig would turn this memory-bound
// kernel VALU-bound, Appendix B).
// One wave per (token, head): lane l rotates the pair (l, l+dh/2), dh=2*WAVE.
__global__ void rope_qk_kernel(__hip_bfloat16* __restrict__ q,
                               __hip_bfloat16* __restrict__ k,
                               const float* __restrict__ cos_sin,
                               const int32_t* __restrict__ positions,
                               const int hq, const int hk, const int dh) {
  const int64_t token = blockIdx.x;
  const int wid = threadIdx.x / WAVE;      // head slot within the block
  const int lane = threadIdx.x % WAVE;
  const int nwave = blockDim.x / WAVE;
  const int half = dh / 2;
  const int pos = positions[token];
  const float* cs = cos_sin + (int64_t)pos * dh;

  for (int h = wid; h < hq + hk; h += nwave) {
    __hip_bfloat16* base = (h < hq) ? q + (token * hq + h) * dh
                                    : k + (token * hk + (h - hq)) * dh;
    for (int i = lane; i < half; i += WAVE) {
      const float c = cs[i], s = cs[half + i];
      const float x0 = bf2f(base[i]);
      const float x1 = bf2f(base[half + i]);
      base[i] = f2bf(x0 * c - x1 * s);
      base[half + i] = f2bf(x1 * c + x0 * s);
    }
  }
}

// ---------------------------------------------------------------- kv_write

// Paged KV cache layout: [num_pages, Hkv, PAGE, dh] so one (page, head) is a
// contiguous PAGE*dh*2B run — the decode kernel streams it line-aligned.
__global__ void kv_write_kernel(__hip_bfloat16* __restrict__ kcache,
                                __hip_bfloat16* __restrict__ vcache,
                                const __hip_bfloat16* __restrict__ k,
                                const __hip_bfloat16* __restrict__ v,
                                const int32_t* __restrict__ slots,
                                const int hkv, const int dh, const int page) {
  const int64_t token = blockIdx.x;
  const int slot = slots[token];
  if (slot < 0) return;  // padding
  const int p = slot / page, off = slot % page;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int nwave = blockDim.x / WAVE;
  const int nvec = dh / 8;  // bf16x8 chunks per row

  for (int h = wid; h < hkv; h += nwave) {
    const bf16x8* ks = reinterpret_cast<const bf16x8*>(k + (token * hkv + h) * dh);
    const bf16x8* vs = reinterpret_cast<const bf16x8*>(v + (token * hkv + h) * dh);
    int64_t dst = (((int64_t)p * hkv + h) * page + off) * dh;
    bf16x8* kd = reinterpret_cast<bf16x8*>(kcache + dst);
    bf16x8* vd = reinterpret_cast<bf16x8*>(vcache + dst);
    for (int i = lane; i < nvec; i += WAVE) {
      kd[i] = ks[i];
      vd[i] = vs[i];
    }
  }
}

// ---------------------------------------------------------- rope_qkv_write

// Fused epilogue of the QKV projection: rotate Q and K heads in place inside
// the packed [T, (Hq+2*Hkv)*dh] GEMM output, and scatter the K/V heads into
// the paged cache — one launch replacing {3x contiguous-copy, rope, kv_write}
// (the .contiguous() copies were 130 ms + 36 ms copyBuffer in the r01
// profile, profiles/r01_bench_llama8b_kernelstats.md).
__global__ void rope_qkv_write_kernel(
    __hip_bfloat16* __restrict__ qkv,        // [T, (hq+2*hkv)*dh]
    __hip_bfloat16* __restrict__ kcache,     // [P, hkv, page, dh]
    __hip_bfloat16* __restrict__ vcache,
    const float* __restrict__ cos_sin,       // [max_pos, dh]
    const int32_t* __restrict__ positions,   // [T]
    const int32_t* __restrict__ slots,       // [T]
    const int hq, const int hkv, const int dh, const int page) {
  const int64_t token = blockIdx.x;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int nwave = blockDim.x / WAVE;
  const int half = dh / 2;
  const int W = (hq + 2 * hkv) * dh;
  const int pos = positions[token];
  const int slot = slots[token];
  const float* cs = cos_sin + (int64_t)pos * dh;
  __hip_bfloat16* row = qkv + token * (int64_t)W;

  // jobs 0..hq-1: rope Q head; hq..hq+hkv-1: rope K head + cache write;
  // hq+hkv..hq+2*hkv-1: V cache write.  grid.y spreads the jobs exactly
  // like the slab twin below (T x 3 blocks for llama-8b): the
  // single-block-per-token form measured 16.2 us in-bench vs the
  // twin's 8.4 on comparable work — launch-parallelism bound.
  // dh == 2*WAVE: lane owns the element pair (lane, lane+64).
  for (int job = blockIdx.y * nwave + wid; job < hq + 2 * hkv;
       job += gridDim.y * nwave) {
    __hip_bfloat16* base = row + job * dh;
    float v0 = bf2f(base[lane]);
    float v1 = bf2f(base[half + lane]);
    if (job < hq + hkv) {
      const float c = cs[lane], sn = cs[half + lane];
      const float r0 = v0 * c - v1 * sn;
      const float r1 = v1 * c + v0 * sn;
      v0 = r0;
      v1 = r1;
      base[lane] = f2bf(v0);
      base[half + lane] = f2bf(v1);
    }
    if (job >= hq && slot >= 0) {
      const int h = (job - hq) % hkv;
      const bool is_k = job < hq + hkv;
      const int p = slot / page, off = slot % page;
      int64_t dst = (((int64_t)p * hkv + h) * page + off) * dh;
      __hip_bfloat16* d = (is_k ? kcache : vcache) + dst;
      d[lane] = f2bf(v0);
      d[half + lane] = f2bf(v1);
    }
  }
}

// slab-fused variant: qkv = sum_sk(work) cast bf16, then rope + cache
// write in the same launch (the split-K reduce of the QKV projection rides
// the epilogue instead of its own kernel)
__global__ void rope_qkv_write_slab_kernel(
    __hip_bfloat16* __restrict__ qkv,        // [T, W] out
    const float* __restrict__ work,          // [sk, Mp, W] slabs
    __hip_bfloat16* __restrict__ kcache,
    __hip_bfloat16* __restrict__ vcache,
    const float* __restrict__ cos_sin,
    const int32_t* __restrict__ positions,
    const int32_t* __restrict__ slots,
    const int hq, const int hkv, const int dh, const int page,
    const int Mp, const int splitk) {
  const int64_t token = blockIdx.x;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int nwave = blockDim.x / WAVE;
  const int half = dh / 2;
  const int W = (hq + 2 * hkv) * dh;
  const int pos = positions[token];
  const int slot = slots[token];
  const float* cs = cos_sin + (int64_t)pos * dh;
  __hip_bfloat16* row = qkv + token * (int64_t)W;

  // grid.y spreads the head-jobs: block (x=token, y) covers jobs
  // [y*nwave .. ) striding gridDim.y*nwave — T x 3 blocks for llama-8b
  // instead of T (the single-block version was launch-latency bound)
  for (int job = blockIdx.y * nwave + wid; job < hq + 2 * hkv;
       job += gridDim.y * nwave) {
    const int col0 = job * dh;
    // reduce this head's slab columns; lane l owns elements (l, l+64)
    float v0 = 0.f, v1 = 0.f;
    for (int sk = 0; sk < splitk; ++sk) {
      const float* slab = work + ((int64_t)sk * Mp + token) * W + col0;
      v0 += slab[lane];
      v1 += slab[half + lane];
    }
    if (job < hq + hkv) {
      const float c = cs[lane], sn = cs[half + lane];
      const float r0 = v0 * c - v1 * sn;
      const float r1 = v1 * c + v0 * sn;
      v0 = r0;
      v1 = r1;
    }
    row[col0 + lane] = f2bf(v0);
    row[col0 + half + lane] = f2bf(v1);
    if (job >= hq && slot >= 0) {
      const int h = (job - hq) % hkv;
      const bool is_k = job < hq + hkv;
      const int p = slot / page, off = slot % page;
      int64_t dst = (((int64_t)p * hkv + h) * page + off) * dh;
      __hip_bfloat16* d = (is_k ? kcache : vcache) + dst;
      d[lane] = f2bf(v0);
      d[half + lane] = f2bf(v1);
    }
  }
}

// slab-fused variant of fused_add_rmsnorm: residual += sum_sk(slabs);
// hidden = rmsnorm(residual) — the split-K reduce of a row-parallel
// projection (o / down) fused into the consuming norm.
template <int BLOCK>
__global__ void fused_add_rmsnorm_slab_kernel(
    __hip_bfloat16* __restrict__ hidden,     // [T, d] out (the norm)
    __hip_bfloat16* __restrict__ residual,   // [T, d] in/out
    const float* __restrict__ work,          // [sk, Mp, d]
    const __hip_bfloat16* __restrict__ weight,
    const int d, const float eps, const int Mp, const int splitk) {
  __shared__ float red[BLOCK / WAVE];
  const int64_t row = blockIdx.x;
  __hip_bfloat16* h = hidden + row * d;
  __hip_bfloat16* r = residual + row * d;

  float ss = 0.f;
  const int nvec = d / 8;
  bf16x8* rv = reinterpret_cast<bf16x8*>(r);
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    bf16x8 b = rv[i];
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = bf2f(b.v[j]);
    for (int sk = 0; sk < splitk; ++sk) {
      const float4* wp = reinterpret_cast<const float4*>(
          work + ((int64_t)sk * Mp + row) * d + i * 8);
      const float4 w0 = wp[0], w1 = wp[1];
      acc[0] += w0.x; acc[1] += w0.y; acc[2] += w0.z; acc[3] += w0.w;
      acc[4] += w1.x; acc[5] += w1.y; acc[6] += w1.z; acc[7] += w1.w;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      b.v[j] = f2bf(acc[j]);
      const float fr = bf2f(b.v[j]);  // norm over the bf16-rounded sum
      ss += fr * fr;
    }
    rv[i] = b;
  }
  ss = block_sum<BLOCK>(ss, red);
  const float inv = rsqrtf(ss / d + eps);
  const bf16x8* wv = reinterpret_cast<const bf16x8*>(weight);
  bf16x8* hv = reinterpret_cast<bf16x8*>(h);
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    bf16x8 b = rv[i], w = wv[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = f2bf(bf2f(b.v[j]) * inv * bf2f(w.v[j]));
    hv[i] = o;
  }
}

// ------------------------------------------------------------- host glue

#include <torch/extension.h>
#include <cstdlib>
#include <ATen/hip/HIPContext.h>

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static int norm_block(int dflt) {
  // per-kernel measured defaults (r02 closing A/Bs): fused_add 512
  // (4.18 vs 4.29 us); the slab twin 256 (8.34 vs 8.88 — its fp32 slab
  // reads prefer more blocks in flight); plain rmsnorm a wash (3.72 vs
  // 3.76), kept at its winner 256.  Env overrides all three for A/B.
  const char* bv = getenv("CALF_RMSNORM_BLOCK");
  return bv ? atoi(bv) : dflt;
}

void rmsnorm(torch::Tensor out, torch::Tensor inp, torch::Tensor weight,
             double eps) {
  TORCH_CHECK(inp.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(inp.scalar_type() == torch::kBFloat16);
  const int d = inp.size(-1);
  TORCH_CHECK(d % 8 == 0, "hidden dim must be a multiple of 8");
  const int64_t rows = inp.numel() / d;
#define LRN(B)                                                                hipLaunchKernelGGL(rmsnorm_kernel<B>, dim3(rows), dim3(B), 0,                                  cur_stream(), (__hip_bfloat16*)out.data_ptr(),                              (const __hip_bfloat16*)inp.data_ptr(),                                      (const __hip_bfloat16*)weight.data_ptr(), d, (float)eps)
  switch (norm_block(256)) {
    case 64: LRN(64); break;
    case 128: LRN(128); break;
    case 512: LRN(512); break;
    default: LRN(256); break;
  }
#undef LRN
  HIP_CHECK_LAST();
}

void fused_add_rmsnorm(torch::Tensor hidden, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  TORCH_CHECK(hidden.is_contiguous() && residual.is_contiguous());
  const int d = hidden.size(-1);
  TORCH_CHECK(d % 8 == 0);
  const int64_t rows = hidden.numel() / d;
  TORCH_CHECK(d <= 8192, "fused_add_rmsnorm: d too large");
  // r02 late A/B (us, 100 iters): 512x4096 -> b512 4.18 / b256 4.29 /
  // b64 6.07; 128x8192 -> 3.61 / 4.49 / 7.86.  Fewer-wider phases win;
  // the wave-per-row barrier-free variant LOSES (the parking is HBM
  // round trips, not the block barrier).  Stream floor ~2.5 us.
  const int blk = norm_block(512);
#define LFAR(B)                                                               hipLaunchKernelGGL(fused_add_rmsnorm_kernel<B>, dim3(rows), dim3(B), 0,                        cur_stream(), (__hip_bfloat16*)hidden.data_ptr(),                           (__hip_bfloat16*)residual.data_ptr(),                                       (const __hip_bfloat16*)weight.data_ptr(), d, (float)eps)
  switch (blk) {
    case 64: LFAR(64); break;
    case 128: LFAR(128); break;
    case 512: LFAR(512); break;
    default: LFAR(256); break;
  }
#undef LFAR
  HIP_CHECK_LAST();
}

void silu_mul(torch::Tensor out, torch::Tensor gateup) {
  TORCH_CHECK(out.is_contiguous() && gateup.is_contiguous());
  const int f = out.size(-1);
  TORCH_CHECK(gateup.size(-1) == 2 * f && f % 8 == 0);
  const int64_t rows = out.numel() / f;
  const int nvec = f / 8;
  const int block = 256;
  const int gx = std::min<int64_t>((nvec + block - 1) / block, 64);
  hipLaunchKernelGGL(silu_mul_kernel, dim3(gx, rows), dim3(block), 0,
                     cur_stream(),
                     (__hip_bfloat16*)out.data_ptr(),
                     (const __hip_bfloat16*)gateup.data_ptr(), rows, f);
  HIP_CHECK_LAST();
}

void rope_qk(torch::Tensor q, torch::Tensor k, torch::Tensor cos_sin,
             torch::Tensor positions) {
  // q: [T, Hq, dh], k: [T, Hkv, dh], cos_sin: [max_pos, dh] f32, positions i32
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && cos_sin.is_contiguous());
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32);
  TORCH_CHECK(positions.scalar_type() == torch::kInt32);
  const int T = q.size(0), hq = q.size(1), dh = q.size(2);
  const int hk = k.size(1);
  TORCH_CHECK(dh == 2 * WAVE, "rope kernel assumes head_dim == 128");
  const int block = 256;
  hipLaunchKernelGGL(rope_qk_kernel, dim3(T), dim3(block), 0, cur_stream(),
                     (__hip_bfloat16*)q.data_ptr(),
                     (__hip_bfloat16*)k.data_ptr(),
                     (const float*)cos_sin.data_ptr(),
                     (const int32_t*)positions.data_ptr(), hq, hk, dh);
  HIP_CHECK_LAST();
}

void rope_qkv_write(torch::Tensor qkv, torch::Tensor kcache,
                    torch::Tensor vcache, torch::Tensor cos_sin,
                    torch::Tensor positions, torch::Tensor slots,
                    int64_t hq, int64_t hkv) {
  TORCH_CHECK(qkv.is_contiguous() && kcache.is_contiguous());
  TORCH_CHECK(positions.scalar_type() == torch::kInt32);
  TORCH_CHECK(slots.scalar_type() == torch::kInt32);
  const int T = qkv.size(0);
  const int page = kcache.size(2);
  const int dh = kcache.size(3);
  TORCH_CHECK(dh == 2 * WAVE, "rope_qkv_write assumes head_dim == 128");
  TORCH_CHECK(qkv.size(1) == (hq + 2 * hkv) * dh, "bad qkv width");
  const int jobs = (int)(hq + 2 * hkv);
  hipLaunchKernelGGL(rope_qkv_write_kernel, dim3(T, (jobs + 3) / 4), dim3(256),
                     0, cur_stream(),
                     (__hip_bfloat16*)qkv.data_ptr(),
                     (__hip_bfloat16*)kcache.data_ptr(),
                     (__hip_bfloat16*)vcache.data_ptr(),
                     (const float*)cos_sin.data_ptr(),
                     (const int32_t*)positions.data_ptr(),
                     (const int32_t*)slots.data_ptr(), (int)hq, (int)hkv, dh,
                     page);
  HIP_CHECK_LAST();
}

void rope_qkv_write_slab(torch::Tensor qkv, torch::Tensor work,
                         torch::Tensor kcache, torch::Tensor vcache,
                         torch::Tensor cos_sin, torch::Tensor positions,
                         torch::Tensor slots, int64_t hq, int64_t hkv,
                         int64_t Mp, int64_t splitk) {
  TORCH_CHECK(qkv.is_contiguous() && work.is_contiguous());
  const int T = qkv.size(0);
  const int page = kcache.size(2);
  const int dh = kcache.size(3);
  TORCH_CHECK(dh == 2 * WAVE);
  const int jobs = (int)(hq + 2 * hkv);
  hipLaunchKernelGGL(rope_qkv_write_slab_kernel, dim3(T, (jobs + 3) / 4),
                     dim3(256), 0, cur_stream(),
                     (__hip_bfloat16*)qkv.data_ptr(),
                     (const float*)work.data_ptr(),
                     (__hip_bfloat16*)kcache.data_ptr(),
                     (__hip_bfloat16*)vcache.data_ptr(),
                     (const float*)cos_sin.data_ptr(),
                     (const int32_t*)positions.data_ptr(),
                     (const int32_t*)slots.data_ptr(), (int)hq, (int)hkv, dh,
                     page, (int)Mp, (int)splitk);
  HIP_CHECK_LAST();
}

void fused_add_rmsnorm_slab(torch::Tensor hidden, torch::Tensor residual,
                            torch::Tensor work, torch::Tensor weight,
                            double eps, int64_t Mp, int64_t splitk) {
  TORCH_CHECK(hidden.is_contiguous() && residual.is_contiguous());
  const int d = hidden.size(-1);
  TORCH_CHECK(d % 8 == 0);
  const int64_t rows = hidden.numel() / d;
#define LFS(B)                                                                hipLaunchKernelGGL(fused_add_rmsnorm_slab_kernel<B>, dim3(rows), dim3(B),                      0, cur_stream(), (__hip_bfloat16*)hidden.data_ptr(),                        (__hip_bfloat16*)residual.data_ptr(),                                       (const float*)work.data_ptr(),                                              (const __hip_bfloat16*)weight.data_ptr(), d, (float)eps,                     (int)Mp, (int)splitk)
  switch (norm_block(256)) {
    case 64: LFS(64); break;
    case 128: LFS(128); break;
    case 512: LFS(512); break;
    default: LFS(256); break;
  }
#undef LFS
  HIP_CHECK_LAST();
}

void kv_write(torch::Tensor kcache, torch::Tensor vcache, torch::Tensor k,
              torch::Tensor v, torch::Tensor slots) {
  // kcache/vcache: [num_pages, Hkv, PAGE, dh]; k/v: [T, Hkv, dh]; slots i32 [T]
  TORCH_CHECK(kcache.is_contiguous() && k.is_contiguous());
  TORCH_CHECK(slots.scalar_type() == torch::kInt32);
  const int T = k.size(0), hkv = k.size(1), dh = k.size(2);
  const int page = kcache.size(2);
  TORCH_CHECK(kcache.size(1) == hkv && kcache.size(3) == dh);
  hipLaunchKernelGGL(kv_write_kernel, dim3(T), dim3(256), 0, cur_stream(),
                     (__hip_bfloat16*)kcache.data_ptr(),
                     (__hip_bfloat16*)vcache.data_ptr(),
                     (const __hip_bfloat16*)k.data_ptr(),
                     (const __hip_bfloat16*)v.data_ptr(),
                     (const int32_t*)slots.data_ptr(), hkv, dh, page);
  HIP_CHECK_LAST();
}
