// Paged-KV GQA decode attention for MI355X (gfx950).
//
// Shape: one query token per sequence (the decode step), grouped-query
// attention against a paged KV cache laid out [num_pages, Hkv, KVPAGE, dh]
// with KVPAGE == 64 so one (page, head) panel is a contiguous 16 KB run the
// wave streams line-aligned.
//
// Decomposition: one workgroup per (sequence, kv_head); NW waves stride the
// sequence's pages.  NW is a template knob: agent decode lives at 2-4
// pages per sequence, where 4-wave blocks idle 1-2 wave slots per CU —
// 2-wave blocks pack twice as many blocks per CU and keep every slot hot
// (env CALF_ATTN_NW overrides; default set by the r31 A/B).  Within a wave, LANE == TOKEN for the QK phase (each
// lane owns one key row, so the G per-head scores are lane-local dots — no
// cross-lane reduction), and LANE == DIM-PAIR for the PV phase (p broadcast
// by shuffle, V row read coalesced 4 B/lane).  Online softmax per wave with
// a final cross-wave m/l combine in LDS.
//
// The kernel is memory-bound by construction (~16 B of KV per lane-op);
// the design targets the HBM3E stream, not MFMA (decode reads each KV byte
// exactly once — matrix cores have nothing to multiply twice).
//
// r02 occupancy A/B: forcing 5 waves/SIMD via __launch_bounds__(.., 5)
// squeezes 102 -> 96 VGPR at 28 B/lane scratch spill — same-box bench
// 241.8 -> 231.8/235.2 and the shuffled-page micro 122.8 -> 177.4 us.
// The spill costs more than the extra wave hides; stay at 4 waves.

#include "common.h"

#define KVPAGE 64
#define HD 128

template <int G, int NW, bool PFETCH>
__global__ __launch_bounds__(NW * WAVE) void attn_decode4_kernel(
    __hip_bfloat16* __restrict__ out,
    const __hip_bfloat16* __restrict__ q,
    const __hip_bfloat16* __restrict__ kcache,
    const __hip_bfloat16* __restrict__ vcache,
    const int32_t* __restrict__ page_table,
    const int32_t* __restrict__ seq_lens,
    const int hq, const int hkv, const int max_pages, const float scale,
    const int q_stride) {   // row stride of q in elements (strided qkv view)
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int slen = seq_lens[seq];
  const int npages = (slen + KVPAGE - 1) / KVPAGE;

  __shared__ float q_lds[G][HD];
  __shared__ float o_lds[NW][G][HD];
  __shared__ float ml_lds[NW][G][2];

  for (int i = threadIdx.x; i < G * HD; i += NW * WAVE) {
    const int g = i / HD, d = i % HD;
    q_lds[g][d] =
        bf2f(q[(int64_t)seq * q_stride + (kvh * G + g) * HD + d]) * scale;
  }
  __syncthreads();

  float m[G], l[G], acc[G][4];  // lane owns dims [4*((lane)%32)... see below
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -INFINITY;
    l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[g][j] = 0.f;
  }
  // dim mapping for PV: HD=128 dims; 64 lanes x 4 dims each covers 256 — so
  // only lanes 0..31 own dims (4*lane..4*lane+3); lanes 32..63 mirror them
  // for the SECOND half of the tokens to double PV throughput.
  const int dlane = lane & 31;        // dim group
  const int thalf = lane >> 5;        // 0: tokens 0..31, 1: tokens 32..63

  const int64_t tbl = (int64_t)seq * max_pages;
  // software-pipelined page ids: the table read is a dependent scalar load
  // (L2/HBM latency) feeding every K/V address — prefetch one iteration
  // ahead so the next page's id is in flight during this page's stream
  int page_next = PFETCH && (wid < npages) ? page_table[tbl + wid] : 0;
  for (int pi = wid; pi < npages; pi += NW) {
    // bar the compiler from hoisting the per-page q_lds re-reads out of
    // this loop: without it LICM keeps G*128 floats live across pages
    // (251-747 VGPR spills measured; see the r01/r03 kernel profiles)
    asm volatile("" ::: "memory");
    const int page = PFETCH ? page_next : page_table[tbl + pi];
    if (PFETCH && pi + NW < npages) page_next = page_table[tbl + pi + NW];
    const int64_t base = (((int64_t)page * hkv + kvh) * KVPAGE) * HD;
    const int token = pi * KVPAGE + lane;
    const bool live = token < slen;

    float s[G];
#pragma unroll
    for (int g = 0; g < G; ++g) s[g] = 0.f;
    {
      // all 16 row loads issue together (one memory round trip), then the
      // FMA sweep consumes them; q re-read from LDS as float4 broadcasts
      // (cheap; every lane reads the same address). Masked lanes clamp to
      // row 0 of the page (valid memory) and overwrite s below.
      const bf16x8* krow = reinterpret_cast<const bf16x8*>(
          kcache + base + (live ? lane : 0) * HD);
      bf16x8 kreg[HD / 8];
#pragma unroll
      for (int i = 0; i < HD / 8; ++i) kreg[i] = krow[i];
#pragma unroll
      for (int i = 0; i < HD / 8; ++i) {
#pragma unroll
        for (int jj = 0; jj < 2; ++jj) {
#pragma unroll
          for (int g = 0; g < G; ++g) {
            const float4 qv =
                *reinterpret_cast<const float4*>(&q_lds[g][i * 8 + jj * 4]);
            s[g] = fmaf(qv.x, bf2f(kreg[i].v[jj * 4 + 0]), s[g]);
            s[g] = fmaf(qv.y, bf2f(kreg[i].v[jj * 4 + 1]), s[g]);
            s[g] = fmaf(qv.z, bf2f(kreg[i].v[jj * 4 + 2]), s[g]);
            s[g] = fmaf(qv.w, bf2f(kreg[i].v[jj * 4 + 3]), s[g]);
          }
        }
      }
    }
    if (!live) {
#pragma unroll
      for (int g = 0; g < G; ++g) s[g] = -INFINITY;
    }

    float p[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float mt = wave_max(s[g]);
      const float mn = fmaxf(m[g], mt);
      const float r = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
      p[g] = (s[g] == -INFINITY) ? 0.f : __expf(s[g] - mn);
      l[g] = l[g] * r + wave_sum(p[g]);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[g][j] *= r;
      m[g] = mn;
    }

    // PV: lanes 0..31 take tokens 0..31, lanes 32..63 take 32..63; each
    // half-wave covers all 128 dims (32 lanes x 4 dims).  V row read is
    // 8 B/lane coalesced within the half's 32 lanes.  ALL 32 rows prefetch
    // up front (32 independent loads in flight) — the serial
    // load->shuffle->fma version was latency-bound at 64.8 us/call
    // (profiles/r01_bench_llama8b_kernelstats.md).
    const int t0 = thalf * 32;
    const int nt = min(32, slen - pi * KVPAGE - t0);
    const float2* vbase = reinterpret_cast<const float2*>(vcache + base) + dlane;
    if (nt > 0) {
      // 2x8-row double buffer, macro-expanded so every array index is a
      // compile-time constant (pointer selection would demote to scratch).
      // (An early-issue variant that hoisted the first 16 V rows above the
      // softmax spilled 226 VGPRs and ran 7x slower — r02; kreg + vA/vB
      // must not be live at once.)
      float2 vA[8], vB[8];
#define PV_LOAD(BUF, C)                                                     \
  _Pragma("unroll") for (int j = 0; j < 8; ++j)                             \
      BUF[j] = vbase[(t0 + min((C) * 8 + j, nt - 1)) * (HD / 4)];
#define PV_FMA(BUF, C)                                                      \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                           \
    const int t = (C) * 8 + j;                                              \
    if (t < nt) {                                                           \
      const __hip_bfloat16* vb =                                            \
          reinterpret_cast<const __hip_bfloat16*>(&BUF[j]);                 \
      _Pragma("unroll") for (int g = 0; g < G; ++g) {                       \
        const float pg = __shfl(p[g], t0 + t, WAVE);                        \
        acc[g][0] = fmaf(pg, bf2f(vb[0]), acc[g][0]);                       \
        acc[g][1] = fmaf(pg, bf2f(vb[1]), acc[g][1]);                       \
        acc[g][2] = fmaf(pg, bf2f(vb[2]), acc[g][2]);                       \
        acc[g][3] = fmaf(pg, bf2f(vb[3]), acc[g][3]);                       \
      }                                                                     \
    }                                                                       \
  }
      PV_LOAD(vA, 0)
      PV_LOAD(vB, 1)
      PV_FMA(vA, 0)
      PV_LOAD(vA, 2)
      PV_FMA(vB, 1)
      PV_LOAD(vB, 3)
      PV_FMA(vA, 2)
      PV_FMA(vB, 3)
#undef PV_LOAD
#undef PV_FMA
    }
  }

  // ---- combine the two half-wave partials: dims are duplicated across
  // halves, tokens split — a cross-half sum via shfl(lane^32) finishes each
  // wave's O.  m/l are wave-uniform already (wave_max/wave_sum).
#pragma unroll
  for (int g = 0; g < G; ++g) {
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[g][j] += __shfl_xor(acc[g][j], 32, WAVE);
  }

  // ---- cross-wave combine in LDS
  if (lane < 32) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
#pragma unroll
      for (int j = 0; j < 4; ++j) o_lds[wid][g][dlane * 4 + j] = acc[g][j];
      if (dlane == 0) {
        ml_lds[wid][g][0] = m[g];
        ml_lds[wid][g][1] = l[g];
      }
    }
  }
  __syncthreads();

  // wave 0 reduces and writes the output rows
  if (wid == 0 && lane < 32) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float M = -INFINITY;
#pragma unroll
      for (int w = 0; w < NW; ++w) M = fmaxf(M, ml_lds[w][g][0]);
      float L = 0.f, o[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float mw = ml_lds[w][g][0];
        const float sc = (mw == -INFINITY) ? 0.f : __expf(mw - M);
        L += sc * ml_lds[w][g][1];
#pragma unroll
        for (int j = 0; j < 4; ++j) o[j] = fmaf(sc, o_lds[w][g][dlane * 4 + j], o[j]);
      }
      const float inv = (L > 0.f) ? 1.f / L : 0.f;
      __hip_bfloat16* orow = out + ((int64_t)seq * hq + kvh * G + g) * HD;
      bf16x4 pack;
#pragma unroll
      for (int j = 0; j < 4; ++j) pack.v[j] = f2bf(o[j] * inv);
      reinterpret_cast<bf16x4*>(orow)[dlane] = pack;
    }
  }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cstdlib>

void attn_decode(torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
                 torch::Tensor vcache, torch::Tensor page_table,
                 torch::Tensor seq_lens, double scale) {
  // q may be a strided row view into the packed qkv GEMM output
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "q heads must be contiguous");
  TORCH_CHECK(kcache.is_contiguous() && vcache.is_contiguous());
  TORCH_CHECK(page_table.scalar_type() == torch::kInt32);
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32);
  const int B = q.size(0), hq = q.size(1), dh = q.size(2);
  const int hkv = kcache.size(1), page = kcache.size(2);
  const int max_pages = page_table.size(1);
  TORCH_CHECK(dh == HD, "decode kernel assumes head_dim == 128");
  TORCH_CHECK(page == KVPAGE, "decode kernel assumes page_size == 64");
  const int G = hq / hkv;
  const char* nw_env = getenv("CALF_ATTN_NW");
  // r02 closing A/Bs: NW=1 (single wave — no cross-wave combine, no
  // barriers) wins the shuffled-page micro 108.0 vs 122.2 us AND the
  // interleaved same-box bench 1,2,2,1 ladder (244.6/254.6 vs
  // 239.4/240.0, ~+4%), and the 512p/128d long-context bench too
  // (71.0 vs 69.1 turns/s at ~10 pages/seq) — block-level parallelism
  // covers the serial pages.  CALF_ATTN_NW=2/4 keeps the split form.
  const int nw = nw_env ? atoi(nw_env) : 1;
  const char* pf_env = getenv("CALF_ATTN_PFETCH");
  const int pfe = pf_env ? atoi(pf_env) : 0;  // prefetch LOST the r02 A/B (120.8 vs 168.1 us)
  dim3 grid(B, hkv), block(nw * WAVE);
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
#define LAUNCH_NW(GV, NWV)                                                \
  do {                                                                    \
    if (pfe) LAUNCH_NWP(GV, NWV, true);                                   \
    else LAUNCH_NWP(GV, NWV, false);                                      \
  } while (0)
#define LAUNCH_NWP(GV, NWV, PFV)                                          \
  hipLaunchKernelGGL((attn_decode4_kernel<GV, NWV, PFV>), grid, block, 0, s, \
                     (__hip_bfloat16*)out.data_ptr(),                     \
                     (const __hip_bfloat16*)q.data_ptr(),                 \
                     (const __hip_bfloat16*)kcache.data_ptr(),            \
                     (const __hip_bfloat16*)vcache.data_ptr(),            \
                     (const int32_t*)page_table.data_ptr(),               \
                     (const int32_t*)seq_lens.data_ptr(), hq, hkv,        \
                     max_pages, (float)scale, (int)q.stride(0))
#define LAUNCH(GV)                                                        \
  do {                                                                    \
    if (nw == 4) LAUNCH_NW(GV, 4);                                        \
    else if (nw == 2) LAUNCH_NW(GV, 2);                                   \
    else LAUNCH_NW(GV, 1);                                                \
  } while (0)
  switch (G) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 4: LAUNCH(4); break;
    case 8: LAUNCH(8); break;
    default: TORCH_CHECK(false, "unsupported GQA group size ", G);
  }
#undef LAUNCH
#undef LAUNCH_NW
#undef LAUNCH_NWP
  HIP_CHECK_LAST();
}
