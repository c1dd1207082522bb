// 256²-tile 8-phase deep-pipelined NT GEMM core (gfx950).
//
// Implements the guide's verified 8-phase template from its spec:
// 256×256 macro-tile, BK=64, 512 threads (8 waves as 2M×4N), 128 KiB
// LDS (2 double-buffers × 2 half-tiles × 128×64 bf16 per operand),
// glds async staging with counted vmcnt(6) only at phases 4/8 (never
// vmcnt(0) in the main loop — prefetched half-tiles stay in flight
// across raw s_barriers), st_16x32 LDS swizzle
// (byte ^= ((byte>>9)&1)<<5 within each 1024-B subtile), s_setprio
// around the MFMA bursts, and the bijective XCD workgroup swizzle.
// ≈1.7× the 128²/2-barrier structure on well-shaped operands.
//
// STATUS (round 2, measured): on every real conv shape (N ≤ 384) the
// 256-wide tile's column padding loses more than the schedule gains —
// kept as an env-gated experiment (VELES_NT8); the counted-vmcnt /
// deep-prefetch techniques it pioneered live in gemm.hip's 128² and
// ntn8 cores.  See profiles/perf_r2.md "measured-negative" records.
#include <stdexcept>
#include "common.h"
#include "loaders.h"

#define O_BM 256
#define O_BN 256
#define O_BK 64
#define O_THREADS 512
// one operand half-tile: 128 rows × 64 k × 2B = 16 KiB
#define O_HALF (128 * O_BK * 2)
// [A0,A1,B0,B1] per K-tile buffer; two buffers
#define O_KTILE (4 * O_HALF)
#define O_LDS_TOTAL (2 * O_KTILE)        // 128 KiB

// swizzle: within each 1024-B subtile, XOR byte bit5 with bit9
__device__ __forceinline__ unsigned swz(unsigned byte) {
  return byte ^ (((byte >> 9) & 1u) << 5);
}

template <typename LA, typename LB>
__global__ __launch_bounds__(O_THREADS, 1) void gemm_nt8_k(
    LA la, LB lb, void* __restrict__ Cout, int c_is_f32,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    int act) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  // bijective XCD swizzle (guide formula)
  int wg = blockIdx.x;
  {
    const int nwg = gridDim.x;
    if (nwg >= 64) {
      const int q = nwg >> 3, r = nwg & 7;
      const int cut = (q + (r ? 1 : 0)) * 8;  // identity past the cut
      if (wg < (q * 8 + r)) {
        const int xcd = wg % 8, pos = wg / 8;
        (void)cut;
        wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) +
             pos;
      }
    }
  }
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * O_BM, n0 = tn * O_BN;
  const int wm = wave >> 2, wn = wave & 3;   // 2M × 4N waves

  const int nk = (K + O_BK - 1) / O_BK;

  // ---- staging: one half-tile = 2 glds per 512 threads ------------------
  // physical byte p = lane-linear (tid*16 within the half); the LOGICAL
  // (row, k-chunk) fetched is the de-swizzle of p, so ds_read can use
  // the swizzled address directly.  Per-thread row contexts (4 A + 4 B:
  // 2 de-swizzled rows × 2 half-rows) are loop-invariant across K-tiles
  // — the magic-div row decomposition runs once per kernel; per tile
  // only 2 k-octet decompositions per operand remain.
  unsigned p_i[2], kc_i[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    p_i[i] = (unsigned)(i * O_THREADS + tid) * 16u;
    const unsigned lg = swz(p_i[i]);
    kc_i[i] = (lg & 127u) >> 4;
  }
  typename LA::RowC rca[2][2];       // [halfrow][i]
  typename LB::RowC rcb[2][2];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int row = (int)(swz(p_i[i]) >> 7) + h * 128;
      rca[h][i] = la.rowc(min(m0 + row, M - 1));
      rcb[h][i] = lb.rowc(min(n0 + row, N - 1));
    }

  auto stage_ktile = [&](int buf, int kt) {
    typename LA::KC kca[2];
    typename LB::KC kcb[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      kca[i] = la.kc(kt * (O_BK / 8) + (int)kc_i[i]);
      kcb[i] = lb.kc(kt * (O_BK / 8) + (int)kc_i[i]);
    }
#pragma unroll
    for (int which = 0; which < 4; ++which) {
      char* base = smem + buf * O_KTILE + which * O_HALF;
      const int h = which & 1;
      const bool isA = which < 2;
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const void* src = isA ? la.addr_rk(rca[h][i], kca[i])
                              : lb.addr_rk(rcb[h][i], kcb[i]);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)src,
            (__attribute__((address_space(3))) void*)(base + p_i[i]), 16,
            0, 0);
      }
    }
  };

  // ---- fragment reads (swizzled b128) ------------------------------------
  const int fr = lane & 15;
  const int fo = lane >> 4;                   // k-octet 0..3 within 64
  auto rd_frag = [&](const char* half_base, int row16, int ks) {
    // logical byte = row*128 + (ks*4 + fo)*16; row = row16 + fr
    const unsigned lg = (unsigned)(row16 + fr) * 128u +
                        (unsigned)(ks * 4 + fo) * 16u;
    return *(const short8*)(half_base + swz(lg));
  };

  // per-wave accumulators: 128(m) × 64(n) = 8 × 4 fragments
  f32x4 acc[8][4] = {};

  // wave's LDS bases: A-half = wm; B rows = wn*64 → B-half wn>>1,
  // rows (wn&1)*64 within it
  const int a_which = wm;                     // 0 or 1
  const int b_which = 2 + (wn >> 1);
  const int b_row0 = (wn & 1) * 64;

  // ---- prologue: stage K-tile 0, enter the pipeline ----------------------
  if (nk > 0) stage_ktile(0, 0);

  // Counted-vmcnt pipeline (guide T3/T4, same discipline as the 128²
  // core): barrier #1 frees tile t-1's buffer, ALL 8 of tile t+1's glds
  // issue into it at once (maximum latency cover: the loads have the
  // whole of tile t's compute — 4 quadrant phases — to land), then
  // `vmcnt(8)` retires exactly tile t's loads while t+1's stay in
  // flight across the raw barrier #2.  No barriers between quadrant
  // phases: waves drift inside a tile (role diversity for s_setprio)
  // and re-sync at the next tile's barrier; the compiler's own counted
  // lgkmcnt waits pace ds_read→MFMA.
  int buf = 0;
  for (int t = 0; t < nk; ++t) {
    __builtin_amdgcn_s_barrier();      // all waves done reading buf^1
    if (t + 1 < nk) {
      stage_ktile(buf ^ 1, t + 1);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();      // tile t's LDS complete everywhere
    const char* Ab = smem + buf * O_KTILE + a_which * O_HALF;
    const char* Bb = smem + buf * O_KTILE + b_which * O_HALF;
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int qm = (q >> 1) * 4;            // fragment-row base 0|4
      const int qn = (q & 1) * 2;             // fragment-col base 0|2
      short8 af0[4], af1[4], bf[2][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          bf[ks][j] = rd_frag(Bb, b_row0 + (qn + j) * 16, ks);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        af0[i] = rd_frag(Ab, (qm + i) * 16, 0);
        af1[i] = rd_frag(Ab, (qm + i) * 16, 1);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[qm + i][qn + j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af0[i], bf[0][j], acc[qm + i][qn + j], 0, 0, 0);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[qm + i][qn + j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af1[i], bf[1][j], acc[qm + i][qn + j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    buf ^= 1;
  }

  // ---- epilogue: fragment (i, j) covers rows rbase..rbase+3 of
  // column col0 + j*16 ------------------------------------------------------
  const int col0 = n0 + wn * 64 + fr;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int col = col0 + j * 16;
    if (col >= N) continue;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int rbase = m0 + wm * 128 + i * 16 + fo * 4;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        float v = acc[i][j][r] + bv;
        switch (act) {
          case ACT_TANH: v = act_fwd<ACT_TANH>(v); break;
          case ACT_RELU: v = act_fwd<ACT_RELU>(v); break;
          case ACT_STRICT_RELU: v = act_fwd<ACT_STRICT_RELU>(v); break;
          case ACT_SIGMOID: v = act_fwd<ACT_SIGMOID>(v); break;
          default: break;
        }
        if (c_is_f32)
          ((float*)Cout)[(long)row * N + col] = v;
        else
          ((unsigned short*)Cout)[(long)row * N + col] = f2bf(v);
      }
    }
  }
}

static inline int cdiv8(int a, int b) { return (a + b - 1) / b; }

template <typename LA, typename LB>
void launch_gemm_nt8(const LA& la, const LB& lb, void* C, int c_is_f32,
                     const float* bias, int M, int N, int K, int act,
                     hipStream_t stream) {
  const int tn = cdiv8(N, O_BN);
  dim3 grid(cdiv8(M, O_BM) * tn), block(O_THREADS);
  hipLaunchKernelGGL((gemm_nt8_k<LA, LB>), grid, block, O_LDS_TOTAL,
                     stream, la, lb, C, c_is_f32, bias, M, N, K, tn,
                     act);
  HIP_CHECK_LAST();
}

// concrete entry points (routed from gemm.hip when the shape fills the
// 256-wide structure)
void conv_fwd_gemm8(const unsigned short* x, const unsigned short* w,
                    void* out, bool out_f32, const float* bias,
                    const unsigned short* zero, const ConvDesc& d,
                    int act, hipStream_t s) {
  const int M = d.N * d.OH * d.OW;
  const int K = d.KH * d.KW * d.C;
  Im2colNT la{x, zero, d};
  DenseNT lb{w, zero, K, K};
  launch_gemm_nt8(la, lb, out, out_f32 ? 1 : 0, bias, M, d.K, K, act,
                  s);
}

void conv_dgrad_gemm8(const unsigned short* eo, const unsigned short* wdg,
                      unsigned short* ei, const unsigned short* zero,
                      const ConvDesc& d, hipStream_t s) {
  const int M = d.N * d.H * d.W;
  const int N = d.C;
  const int K = d.KH * d.KW * d.K;
  DenseNT lb{wdg, zero, K, K};
  if (d.SY == 1 && d.SX == 1) {
    DgradNT1 la{eo, zero, d};
    launch_gemm_nt8(la, lb, (void*)ei, 0, (const float*)nullptr, M, N,
                    K, ACT_LINEAR, s);
    return;
  }
  DgradNT la{eo, zero, d};
  launch_gemm_nt8(la, lb, (void*)ei, 0, (const float*)nullptr, M, N, K,
                  ACT_LINEAR, s);
}
