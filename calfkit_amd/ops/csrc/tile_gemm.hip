// Tiled MFMA GEMM for the serving decode batches (M = 65..512+) on MI355X.
//
// out[M,N] = x[M,K] @ W^T, W stored [N,K] (nn.Linear layout), bf16 in,
// bf16 out (direct) or fp32 split-K slabs (work[sk][Mp][N], the same slab
// contract the skinny path's fused consumers read — rope_qkv_write_slab /
// fused_add_rmsnorm_slab in elementwise.hip).
//
// Why this kernel exists: at the headline batch (512) the
// projections ran hipBLASLt at 0.65-0.84 PF/s — fill/latency-bound
// (profiles/gemm_algo_table.json: qkv 32.8 us, o 26.5, down 72).  The
// shapes have only 32-96 natural 256-wide tiles on a 256-CU chip, so the
// levers are per-block LATENCY via split-K (guide: "(N/64)*SPLITK ~
// 1/2-1x the CU count") and per-CU rate via a counted-vmcnt glds pipeline.
//
// Structure (CDNA4 guide §5 "256^2 8-phase template" adapted to 256xBN):
// - 256xBN tile (BN 128/256), BK=64, 8 waves (512 thr), 2 LDS buffers.
// - Both operands stage global->LDS by __builtin_amdgcn_global_load_lds
//   (16 B/lane), LDS image lane-linear, XOR swizzle on the SOURCE address
//   re-applied on the ds_read_b128 side (guide §5.4 rule 21 / T2:
//   byte ^= (row&7)<<4 -> conflict-free column-slice reads).
// - A K-tile runs as 8 phases (BN=256: m-quadrant x n-half; BN=128: 4
//   phases) of 8 MFMAs behind raw s_barriers with s_setprio(1) around the
//   MFMA cluster (T5).  Wave rows/cols interleave at 16-row granularity
//   (row = mi*16*WR + wr*16) so each phase consumes exactly ONE 128-row
//   stage unit per operand — that is what makes counted waits possible.
// - Next tile's 3-4 stage units issue one-per-phase in first-need order
//   (A0,B0,B1,A1); waits are counted s_waitcnt vmcnt(N) placed before the
//   barrier that precedes the consuming phase's ds_reads — the pipeline
//   never drains (T3+T4; v1 of this kernel drained vmcnt(0) per K-tile
//   via __syncthreads and measured 42% MFU — profiles/r02_tile_gemm.md).
// - XCD-bijective block remap (T1) keeps one n-panel's blocks on one XCD.

#include "common.h"

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;

#define MFMA16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

#define VMWAIT(N) asm volatile("s_waitcnt vmcnt(" #N ")" ::: "memory")

// FLAGS bits: 1 = drop the pre-MFMA barrier; 2 = drop setprio around the
// MFMA cluster (guide worked-example 2: setprio HURT a lockstep GEMM);
// 4 = issue the phase's stage unit BETWEEN the two ks MFMA half-clusters
// (glds issue ~60 cyc among bare MFMAs vs 100-185 in a load phase).
template <int BN, int FLAGS>
__global__ __launch_bounds__(512, 2) void tile_gemm_kernel(
    float* __restrict__ work,             // [splitk, Mp, N] fp32 slabs
    __hip_bfloat16* __restrict__ out,     // [M, N] (direct == 1)
    const __hip_bfloat16* __restrict__ x, // [M, K]
    const __hip_bfloat16* __restrict__ w, // [N, K]
    const int M, const int N, const int K, const int m_tiles,
    const int splitk, const int direct, const int Mp) {
  constexpr int BM = 256, BK = 64;
  constexpr int WC = (BN == 256) ? 4 : 2;  // waves along N
  constexpr int WR = 8 / WC;               // waves along M
  constexpr int MI = BM / 16 / WR;         // m-frags per wave (8 / 4)
  constexpr int NI = BN / 16 / WC;         // n-frags per wave (4)
  constexpr int A_ELEMS = BM * BK;
  constexpr int B_ELEMS = BN * BK;
  constexpr int BUF = A_ELEMS + B_ELEMS;
  // FLAGS bit 16: 3-deep LDS ring (BN=128 fits 3 x 48 KiB in 160 KiB) —
  // units stage TWO tiles ahead, so every load has ~2 tiles of flight
  constexpr int BUFS = (BN == 128 && (FLAGS & 16)) ? 3 : 2;
  __shared__ __attribute__((aligned(16))) __hip_bfloat16 smem[BUFS * BUF];

  // ---- block id -> (n_tile, m_tile, sk) with XCD-bijective remap (T1)
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
  const int wr = wid / WC, wc = wid % WC;
  // staging: thread t covers image row sr (8 rows per wave per call),
  // 16B chunk sc of the 128 B row; source chunk is XOR-swizzled (rule 21)
  const int sr = tid / 8;  // == wid*8 + lane/8
  const int sc = tid % 8;

  // one stage unit = 128 image rows = 2 glds per wave (16 KiB total)
  const auto stage_a = [&](int buf, int ktile, int half) {
    const int kk = k0 + ktile * BK;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int rloc = half * 128 + c * 64 + sr;
      const int grow = min(m0 + rloc, M - 1);  // tail tile row clamp
      const int chunk = sc ^ (rloc & 7);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(
              x + (int64_t)grow * K + kk + chunk * 8),
          reinterpret_cast<unsigned int*>(
              &smem[buf * BUF + (half * 128 + c * 64 + wid * 8) * BK]),
          16, 0, 0);
    }
  };
  const auto stage_b = [&](int buf, int ktile, int half) {
    const int kk = k0 + ktile * BK;
#pragma unroll
    for (int c = 0; c < 2; ++c) {  // one unit = 128 rows = 2 x 64-row calls
      const int rloc = half * 128 + c * 64 + sr;
      const int chunk = sc ^ (rloc & 7);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(
              w + (int64_t)(n0 + rloc) * K + kk + chunk * 8),
          reinterpret_cast<unsigned int*>(
              &smem[buf * BUF + A_ELEMS + (half * 128 + c * 64 + wid * 8) * BK]),
          16, 0, 0);
    }
  };
  // BN=128's B unit is 2 glds over rows 0..127 via c loop; half==0 always.
  // fragment reads: 16-lane groups span 16 rows at one k-chunk -> the T2
  // column-slice pattern; same XOR as the staged source.  Wave rows/cols
  // interleave so mi<MI/2 lies in stage half 0.
  const auto read_a = [&](int buf, int mi, int ks) {
    const int row = mi * (16 * WR) + wr * 16 + l16;
    const int chunk = (ks * 4 + l4) ^ (row & 7);
    return *reinterpret_cast<const bf16x8v*>(
        &smem[buf * BUF + row * BK + chunk * 8]);
  };
  const auto read_b = [&](int buf, int ni, int ks) {
    const int row = ni * (16 * WC) + wc * 16 + l16;
    const int chunk = (ks * 4 + l4) ^ (row & 7);
    return *reinterpret_cast<const bf16x8v*>(
        &smem[buf * BUF + A_ELEMS + row * BK + chunk * 8]);
  };

  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // prologue: tile 0 in first-need order; counted waits below cover it
  if (BN == 256) {
    stage_b(0, 0, 0);
    stage_b(0, 0, 1);
    stage_a(0, 0, 0);
    stage_a(0, 0, 1);
  } else {
    stage_b(0, 0, 0);
    stage_a(0, 0, 0);
    stage_a(0, 0, 1);
  }

  if (BUFS == 3 && nt > 1) {  // deep ring: tile 1 prefetches up front too
    stage_b(1, 1, 0);
    stage_a(1, 1, 0);
    stage_a(1, 1, 1);
  }

  // FLAGS bit 64 (T5 static form): the second-dispatched half of the
  // workgroup loses VALU arbitration on every segment; ONE static
  // priority raise for it (provably wave-uniform via readfirstlane)
  if constexpr ((FLAGS & 64) != 0) {
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
      __builtin_amdgcn_s_setprio(1);
  }

  // pre-loop: tile 0's units landed, all waves synced before any LDS read
  if (BUFS == 3) {
    if (nt > 1) {
      VMWAIT(6);
    } else {
      VMWAIT(0);
    }
  } else {
    VMWAIT(2);
  }
  __builtin_amdgcn_s_barrier();

  int cur = 0;
  for (int t = 0; t < nt; ++t) {
    const bool pf = t + 1 < nt;
    const int nb = BUFS == 3 ? (t + 2) % 3 : (cur ^ 1);
    const bool pf2 = BUFS == 3 ? (t + 2 < nt) : pf;
    // Phase body order matters (PMC r02: stage-issue at phase head cost
    // ~300 cyc/phase of MFMA-pipe idle): ds_reads FIRST, then the stage
    // issue, then barrier -> MFMA cluster.  Counted waits sit at a
    // phase's TAIL (before its end barrier) covering the NEXT phase's
    // units, so the barrier orders every wave's wait before any read.
    if (BN == 256) {
      // 4 phases of 16 MFMA (one m-quadrant x full BN); stage units of
      // t+1 issue one-per-phase in first-need order B0,B1,A0,A1.
      bf16x8v bfr[NI][2];
      // FLAGS bit 32: A-fragment double-buffer — phase p+1's ds_reads
      // issue BEFORE phase p's MFMA cluster, so their ~50-100 cyc LDS
      // latency hides under the MFMAs instead of parking the wave at the
      // next phase head (r02 PMC: 37.6% of wave time parked).
      bf16x8v af0[2][2], af1[2][2];
#pragma unroll
      for (int ph = 0; ph < 4; ++ph) {
        bf16x8v(&afr)[2][2] = (ph & 1) ? af1 : af0;
        bf16x8v(&afn)[2][2] = (ph & 1) ? af0 : af1;
        if (ph == 0) {
#pragma unroll
          for (int ni = 0; ni < NI; ++ni)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) bfr[ni][ks] = read_b(cur, ni, ks);
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) afr[mi][ks] = read_a(cur, mi, ks);
        }
        if constexpr ((FLAGS & 32) != 0) {
          // prefetch only WITHIN an A-half: A(ph+1) may be read at ph's
          // head only when it shares ph's stage half (phases 0->1, 2->3);
          // crossing the half boundary would read A1 before its ph1-tail
          // wait (the r02 f47/f63 numerics failures)
          if (ph == 2) {
            // own read first (half1 just landed behind the ph1-tail wait)
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
              for (int ks = 0; ks < 2; ++ks)
                afr[mi][ks] = read_a(cur, ph * 2 + mi, ks);
          }
          if (ph == 0 || ph == 2) {
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
              for (int ks = 0; ks < 2; ++ks)
                afn[mi][ks] = read_a(cur, (ph + 1) * 2 + mi, ks);
          }
        } else if (ph > 0) {
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks)
              afr[mi][ks] = read_a(cur, ph * 2 + mi, ks);
        }
        const auto stage_ph = [&] {
          if (!pf) return;
          if (ph == 0) stage_b(nb, t + 1, 0);
          else if (ph == 1) stage_b(nb, t + 1, 1);
          else if (ph == 2) stage_a(nb, t + 1, 0);
          else stage_a(nb, t + 1, 1);
        };
        if constexpr (!(FLAGS & 4)) stage_ph();
        if constexpr (!(FLAGS & 1)) __builtin_amdgcn_s_barrier();
        if constexpr (!(FLAGS & 2)) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          if (ks == 1 && (FLAGS & 4)) stage_ph();
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < NI; ++ni)
              acc[ph * 2 + mi][ni] =
                  MFMA16(afr[mi][ks], bfr[ni][ks], acc[ph * 2 + mi][ni]);
        }
        if constexpr (!(FLAGS & 2)) __builtin_amdgcn_s_setprio(0);
        if (ph == 1) {
          // ph2 reads A1(t): outstanding {A1, B0', B1'} -> vmcnt(4);
          // final tile issues nothing -> drain what remains
          if (pf) {
            VMWAIT(4);
          } else {
            VMWAIT(0);
          }
          __builtin_amdgcn_s_barrier();
        } else if (ph == 3) {
          if (pf) VMWAIT(2);
          __builtin_amdgcn_s_barrier();
        } else if constexpr (!(FLAGS & 8)) {
          __builtin_amdgcn_s_barrier();
        }
      }
    } else {
      // BN=128: 2 phases of 16 MFMA (2 mi x 4 ni); units B,A0,A1.
      bf16x8v bfr[NI][2];
      bf16x8v af0[2][2], af1[2][2];
#pragma unroll
      for (int ph = 0; ph < 2; ++ph) {
        bf16x8v(&afr)[2][2] = (ph & 1) ? af1 : af0;
        bf16x8v(&afn)[2][2] = (ph & 1) ? af0 : af1;
        if (ph == 0) {
#pragma unroll
          for (int ni = 0; ni < NI; ++ni)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) bfr[ni][ks] = read_b(cur, ni, ks);
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) afr[mi][ks] = read_a(cur, mi, ks);
        }
        if constexpr ((FLAGS & 32) != 0 && BUFS == 3) {
          // safe only with the 3-deep ring: tile t's units landed behind
          // the PREVIOUS tile's wait, so cross-half prefetch cannot race
          if (ph < 1) {
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
              for (int ks = 0; ks < 2; ++ks)
                afn[mi][ks] = read_a(cur, 2 + mi, ks);
          }
        } else if (ph > 0) {
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks)
              afr[mi][ks] = read_a(cur, ph * 2 + mi, ks);
        }
        const int st = BUFS == 3 ? t + 2 : t + 1;
        const auto stage_ph = [&] {
          if (!pf2) return;
          if (ph == 0) {
            stage_b(nb, st, 0);
            stage_a(nb, st, 0);
          } else {
            stage_a(nb, st, 1);
          }
        };
        if constexpr (!(FLAGS & 4)) stage_ph();
        if constexpr (!(FLAGS & 1)) __builtin_amdgcn_s_barrier();
        if constexpr (!(FLAGS & 2)) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          if (ks == 1 && (FLAGS & 4)) stage_ph();
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < NI; ++ni)
              acc[ph * 2 + mi][ni] =
                  MFMA16(afr[mi][ks], bfr[ni][ks], acc[ph * 2 + mi][ni]);
        }
        if constexpr (!(FLAGS & 2)) __builtin_amdgcn_s_setprio(0);
        if (BUFS == 3) {
          // deep ring: tile t's units landed >= 2 tiles ago; the only
          // wait is at ph1's tail, covering tile t+1's whole set
          if (ph == 1 && pf) {
            if (pf2) {
              VMWAIT(6);
            } else {
              VMWAIT(0);
            }
            __builtin_amdgcn_s_barrier();
            continue;
          }
          if (ph == 1) __builtin_amdgcn_s_barrier();
          continue;
        }
        if (ph == 0) {
          // ph1 reads A1(t): outstanding {A1, B', A0'} -> vmcnt(4)
          if (pf) {
            VMWAIT(4);
          } else {
            VMWAIT(0);
          }
        } else if (pf) {
          // next ph0 reads {B',A0'}: outstanding {B',A0',A1'} -> vmcnt(2)
          VMWAIT(2);
        }
        __builtin_amdgcn_s_barrier();
      }
    }
    cur = BUFS == 3 ? (t + 1) % 3 : (cur ^ 1);
  }

  // ---- epilogue.  C layout (mfma_f32_16x16x32): col = l16, row = l4*4+r.
  // Wave rows interleave: m-frag mi -> tile row mi*16*WR + wr*16.
  if (direct) {
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = m0 + mi * (16 * WR) + wr * 16 + l4 * 4 + r;
          if (row < M)
            out[(int64_t)row * N + n0 + ni * (16 * WC) + wc * 16 + l16] =
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
          const int row = m0 + mi * (16 * WR) + wr * 16 + l4 * 4 + r;
          slab[(int64_t)row * N + n0 + ni * (16 * WC) + wc * 16 + l16] =
              acc[mi][ni][r];
        }
  }
}

// ---------------------------------------------------------------------------
// Small-tile variant: 64xBN tiles, 4 waves, 3 blocks/CU.
//
// Why: at M=512 the 256-wide kernel has only 2 m-tiles, so qkv/o land 48-96
// natural tiles on a 256-CU chip and need split-K (and its fp32 slab tax) to
// fill it; the library's winning macro-tiles on exactly these shapes are
// MT96x128/MT128x64 (profiles/r02_bench_kernelstats.md).  BM=64 gives 8
// m-tiles -> 256-384 natural tiles with NO split-K, and the latency-hiding
// burden moves from intra-block pipelining (8-phase counted-wait ring) to
// occupancy: 3 blocks/CU x 4 waves = 12 waves/CU, so one block's post-stage
// drain parks under another block's MFMAs.  B panels are re-read M/64 times,
// but consecutive logical blocks share an n-panel and the XCD-bijective remap
// keeps them on one XCD, so re-reads hit that XCD's 4 MB L2 (panel = 1 MB at
// BN=128, K-slice resident) instead of HBM.
// BUFS=2: per-tile drain (VMWAIT 0), 3 blocks/CU hide it across blocks.
// BUFS=3: ring stages t+2, counted vmcnt(6) tail wait — no drain, but
// 72 KiB LDS caps occupancy at 2 blocks/CU.  CALF_SMALL_BUFS A/Bs them.
template <int BN, int BUFS>
__global__ __launch_bounds__(256, 3) void tile_gemm_small_kernel(
    __hip_bfloat16* __restrict__ out,     // [M, N]
    const __hip_bfloat16* __restrict__ x, // [M, K]
    const __hip_bfloat16* __restrict__ w, // [N, K]
    const int M, const int N, const int K, const int m_tiles) {
  constexpr int BM = 64, BK = 64;
  constexpr int WC = 4;             // waves along N
  constexpr int NI = BN / 16 / WC;  // n-frags per wave (2 at BN=128)
  constexpr int MI = BM / 16;       // m-frags per wave (4); WR == 1
  constexpr int A_ELEMS = BM * BK;
  constexpr int B_ELEMS = BN * BK;
  constexpr int BUF = A_ELEMS + B_ELEMS;
  __shared__ __attribute__((aligned(16))) __hip_bfloat16 smem[BUFS * BUF];

  int bid = blockIdx.x;
  {  // XCD-bijective remap (T1): contiguous logical range per XCD
    const int nwg = gridDim.x, q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, i = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  // m inner: consecutive logical blocks share the n-panel (L2 reuse)
  const int n_t = bid / m_tiles, m_t = bid % m_tiles;
  const int n0 = n_t * BN, m0 = m_t * BM;
  const int nt = K / BK;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE, lane = tid % WAVE;
  const int l16 = lane & 15, l4 = lane >> 4;
  const int wc = wid;  // WR == 1: every wave owns all 64 rows, 32 cols
  const int sr = tid / 8;  // staging: 32 image rows per glds call
  const int sc = tid % 8;

  const auto stage_a = [&](int buf, int ktile) {
    const int kk = ktile * BK;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int rloc = c * 32 + sr;
      const int grow = min(m0 + rloc, M - 1);
      const int chunk = sc ^ (rloc & 7);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(
              x + (int64_t)grow * K + kk + chunk * 8),
          reinterpret_cast<unsigned int*>(
              &smem[buf * BUF + (c * 32 + wid * 8) * BK]),
          16, 0, 0);
    }
  };
  const auto stage_b = [&](int buf, int ktile) {
    const int kk = ktile * BK;
#pragma unroll
    for (int c = 0; c < BN / 32; ++c) {
      const int rloc = c * 32 + sr;
      const int chunk = sc ^ (rloc & 7);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(
              w + (int64_t)(n0 + rloc) * K + kk + chunk * 8),
          reinterpret_cast<unsigned int*>(
              &smem[buf * BUF + A_ELEMS + (c * 32 + wid * 8) * BK]),
          16, 0, 0);
    }
  };
  const auto read_a = [&](int buf, int mi, int ks) {
    const int row = mi * 16 + l16;
    const int chunk = (ks * 4 + l4) ^ (row & 7);
    return *reinterpret_cast<const bf16x8v*>(
        &smem[buf * BUF + row * BK + chunk * 8]);
  };
  const auto read_b = [&](int buf, int ni, int ks) {
    const int row = ni * (16 * WC) + wc * 16 + l16;
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
  if (BUFS == 3 && nt > 1) {
    stage_a(1, 1);
    stage_b(1, 1);
    VMWAIT(6);
  } else {
    VMWAIT(0);
  }
  __builtin_amdgcn_s_barrier();

  int cur = 0;
  for (int t = 0; t < nt; ++t) {
    const bool pf = t + 1 < nt;
    const int st = BUFS == 3 ? t + 2 : t + 1;
    const bool pf2 = st < nt;
    const int nb = BUFS == 3 ? (t + 2) % 3 : (cur ^ 1);
    bf16x8v afr[MI][2], bfr[NI][2];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) afr[mi][ks] = read_a(cur, mi, ks);
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) bfr[ni][ks] = read_b(cur, ni, ks);
    // next stages issue behind this tile's MFMAs; the tail wait +
    // barrier land tile t+1 before the next head's ds_reads.  nb was
    // last READ at t-1 (BUFS=2) / t-1's ring slot (BUFS=3), and every
    // wave passed the t-1 tail barrier after those reads.
    if (pf2) {
      stage_a(nb, st);
      stage_b(nb, st);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = MFMA16(afr[mi][ks], bfr[ni][ks], acc[mi][ni]);
    if (pf) {
      // BUFS=3: t+1's 6 units landed when t+2's 6 are still in flight
      if (BUFS == 3 && pf2) {
        VMWAIT(6);
      } else {
        VMWAIT(0);
      }
      __builtin_amdgcn_s_barrier();
    }
    cur = BUFS == 3 ? (t + 1) % 3 : (cur ^ 1);
  }

  // epilogue (mfma_f32_16x16x32 C layout: col = l16, row = l4*4 + r)
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + mi * 16 + l4 * 4 + r;
        if (row < M)
          out[(int64_t)row * N + n0 + ni * (16 * WC) + wc * 16 + l16] =
              f2bf(acc[mi][ni][r]);
      }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cstdlib>

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
  const char* fv = getenv("CALF_TILE_FLAGS");
  const int flags = fv ? atoi(fv) : 95;  // 31 + T5 static young-half priority (r02 sweep)
#define LT(BNV, F)                                                           \
  hipLaunchKernelGGL((tile_gemm_kernel<BNV, F>), dim3(blocks), dim3(512), 0, \
                     s, (float*)work.data_ptr(),                             \
                     (__hip_bfloat16*)out.data_ptr(),                        \
                     (const __hip_bfloat16*)x.data_ptr(),                    \
                     (const __hip_bfloat16*)w.data_ptr(), M, N, K, m_tiles,  \
                     (int)splitk, direct, Mp)
#define LTF(BNV)                                                             \
  switch (flags) {                                                           \
    case 1: LT(BNV, 1); break;                                               \
    case 2: LT(BNV, 2); break;                                               \
    case 3: LT(BNV, 3); break;                                               \
    case 4: LT(BNV, 4); break;                                               \
    case 5: LT(BNV, 5); break;                                               \
    case 6: LT(BNV, 6); break;                                               \
    case 7: LT(BNV, 7); break;                                               \
    case 15: LT(BNV, 15); break;                                             \
    case 23: LT(BNV, 23); break;                                             \
    case 31: LT(BNV, 31); break;                                             \
    case 47: LT(BNV, 47); break;                                             \
    case 63: LT(BNV, 63); break;                                             \
    case 95: LT(BNV, 95); break;                                             \
    default: LT(BNV, 0); break;                                              \
  }
  if (bn == 256) {
    LTF(256);
  } else {
    LTF(128);
  }
#undef LTF
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

void tile_gemm_small(torch::Tensor out, torch::Tensor x, torch::Tensor w) {
  // out[M,N] = x[M,K] @ w[N,K]^T via 64x128 tiles — the natural-fill
  // variant for the M=384/512 projections (no split-K, direct bf16 out)
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(N % 128 == 0, "tile_gemm_small: N % 128 != 0");
  TORCH_CHECK(K % 64 == 0, "tile_gemm_small: K % 64 != 0");
  const int m_tiles = (M + 63) / 64;
  const int blocks = (N / 128) * m_tiles;
  hipStream_t s = at::hip::getCurrentHIPStream().stream();
  const char* bv = getenv("CALF_SMALL_BUFS");
  // r02 A/B: shallow K (<=1024, 8-16 K-iters) wins with 2-buf + 3-block
  // occupancy (o70b 12.8 vs 13.7 us); deeper K amortizes the ring's
  // occupancy loss over more drain-free iterations (down8b K=1792:
  // 14.5 vs 15.6; qkv K=4096: 36.2 vs 44.2)
  const int bufs = bv ? atoi(bv) : (K > 1024 ? 3 : 2);
  if (bufs == 3) {
    hipLaunchKernelGGL((tile_gemm_small_kernel<128, 3>), dim3(blocks),
                       dim3(256), 0, s, (__hip_bfloat16*)out.data_ptr(),
                       (const __hip_bfloat16*)x.data_ptr(),
                       (const __hip_bfloat16*)w.data_ptr(), M, N, K, m_tiles);
  } else {
    hipLaunchKernelGGL((tile_gemm_small_kernel<128, 2>), dim3(blocks),
                       dim3(256), 0, s, (__hip_bfloat16*)out.data_ptr(),
                       (const __hip_bfloat16*)x.data_ptr(),
                       (const __hip_bfloat16*)w.data_ptr(), M, N, K, m_tiles);
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
