// MFMA bf16 GEMM cores for gfx950 — the compute heart of veles.znicz_amd.
//
// Replaces the reference's tiled OpenCL/CUDA GEMM template
// (znicz/ocl/matrix_multiplication.cl + all2all.cl/conv.cl includes) with
// two CDNA4-native cores:
//
//   core NT:  C[M,N] = act(A[M,K] · B[N,K]ᵀ + bias)
//     both operands K-innermost (the natural layout of FC fwd, FC dgrad
//     with a persistent transposed-weight copy, and implicit-im2col conv
//     fwd/dgrad).  128×128 macro-tile, BK=64, 256 threads (4 waves in
//     2×2), v_mfma_f32_16x16x32_bf16, double-buffered LDS filled by
//     global_load_lds_dwordx4 (async DMA, 16 B/lane), im2col unrolling
//     done IN the per-lane DMA source address (never materialized in
//     HBM; padding reads a zero page).
//
//   core TN:  C[M,N](fp32) = beta·C + Σ_k A[k][m]·B[k][n]
//     both operands K-major (batch-major) — the wgrad shape for FC and
//     conv (k = N·OH·OW).  Staged by coalesced element loads packed to
//     short8 + swizzled ds_write_b128 (XOR (row&7)<<4 — bank-conflict
//     free), fragments read back with the matching XOR.
//
// Wave64 / LDS-tiled throughout; nothing here is a CUDA translation.
#include <cstdlib>
#include <mutex>
#include <stdexcept>
#include <unordered_map>
#include "common.h"

#define BM 128
#define BN 128
#define BK 64
#define NT_LDS_A (BM * BK * 2)          // 16 KiB
#define NT_LDS_BUF (NT_LDS_A * 2)       // A + B per buffer
#define NT_LDS_TOTAL (NT_LDS_BUF * 2)   // double-buffered = 64 KiB

// ---------------------------------------------------------------------------
// operand loaders (per-lane global source address generators)
// ---------------------------------------------------------------------------

#include "loaders.h"

// ---------------------------------------------------------------------------
// core NT
// ---------------------------------------------------------------------------

template <typename LA, typename LB, typename OUT_T, int ACT>
__global__ __launch_bounds__(256, 2) void gemm_nt_k(
    LA la, LB lb, OUT_T* __restrict__ C, const float* __restrict__ bias,
    const unsigned short* __restrict__ dact, int M, int N, int K,
    int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tile = xcd_tile(blockIdx.x, gridDim.x);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN;
  const int wr = wave >> 1, wc = wave & 1;   // 2x2 waves of 64x64

  // split-K (gridDim.y slices; fp32-out atomic epilogue) fills the
  // chip on short-and-wide FC shapes (batch 256 x N<=4096: 16-64 MN
  // tiles) — bias/activation run in a separate tiny pass.
  const int nk_all = (K + BK - 1) / BK;
  const int sp = gridDim.y, sid = blockIdx.y;
  const int t_lo = (int)(((long)nk_all * sid) / sp);
  const int nk = (int)(((long)nk_all * (sid + 1)) / sp);

  // Slot swizzle: slot s of row r holds k-octet s ^ (r&7) ^ ((r>>3)&1).
  // The glds DMA destination is lane-linear (cannot pad rows), so the
  // b128 fragment reads of 16 consecutive rows would hit 128-B-strided
  // addresses ≡ 0 (mod 64 dw) — an 8-way bank conflict measured at
  // 13-16% of NT wave cycles.  Swizzling WHICH k-octet each slot loads
  // spreads the 16 lanes of a fragment read over distinct banks.
  // Per-thread ROW contexts are loop-invariant across K-tiles: the
  // magic-division row decomposition runs once per kernel here, and
  // per tile only the two k-octet decompositions + add/bounds remain
  // (the staging address math used to dominate the VALU pipe).
  const int rl = lane >> 3;            // row within 8-row chunk
  const int k8 = lane & 7;             // 16-B slot within the 64-k row
  typename LA::RowC rca[4];
  typename LB::RowC rcb[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int row = (wave * 4 + i) * 8 + rl;
    rca[i] = la.rowc(min(m0 + row, M - 1));
    rcb[i] = lb.rowc(min(n0 + row, N - 1));
  }

  auto stage = [&](int buf, int kt) {
    char* As = smem + buf * NT_LDS_BUF;
    char* Bs = As + NT_LDS_A;
    const int k8base = kt * (BK / 8);
    const auto kca0 = la.kc(k8base + (k8 ^ rl));
    const auto kca1 = la.kc(k8base + (k8 ^ rl ^ 1));
    const auto kcb0 = lb.kc(k8base + (k8 ^ rl));
    const auto kcb1 = lb.kc(k8base + (k8 ^ rl ^ 1));
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int chunk = wave * 4 + i;  // 16 chunks x 8 rows = 128 rows
      const int row = chunk * 8 + rl;
      const int k8e = k8 ^ rl ^ (chunk & 1);   // swizzled source octet
      const int kc0 = (k8base + k8e) * 8;
      const bool straddle = (kc0 < K) & (kc0 + 8 > K);
      if constexpr (!LA::MAY_SLOW && !LB::MAY_SLOW) {
        // fast loaders: the only per-element case is the K%8 tail
        if (!straddle) {
          const void* sa = la.addr_rk(rca[i], (i & 1) ? kca1 : kca0);
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) void*)sa,
              (__attribute__((address_space(3))) void*)(As + chunk * 1024),
              16, 0, 0);
          const void* sb = lb.addr_rk(rcb[i], (i & 1) ? kcb1 : kcb0);
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) void*)sb,
              (__attribute__((address_space(3))) void*)(Bs + chunk * 1024),
              16, 0, 0);
        } else {
          unsigned short va[8], vb[8];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            va[j] = la.elem(min(m0 + row, M - 1), kc0 + j);
            vb[j] = lb.elem(min(n0 + row, N - 1), kc0 + j);
          }
          *(short8*)(As + chunk * 1024 + lane * 16) = *(short8*)va;
          *(short8*)(Bs + chunk * 1024 + lane * 16) = *(short8*)vb;
        }
      } else {
        // a packed loader may refuse individual chunks (addr()==nullptr:
        // kh-row crossing / padding / misalignment) — per-operand path
        const void* sa = straddle ? nullptr
                                  : la.addr(min(m0 + row, M - 1),
                                            k8base + k8e);
        const void* sb = straddle ? nullptr
                                  : lb.addr(min(n0 + row, N - 1),
                                            k8base + k8e);
        if (sa) {
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) void*)sa,
              (__attribute__((address_space(3))) void*)(As + chunk * 1024),
              16, 0, 0);
        } else {
          unsigned short va[8];
          if constexpr (LA::MAY_SLOW)
            la.slowload(min(m0 + row, M - 1), kc0, va);
          else {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              va[j] = la.elem(min(m0 + row, M - 1), kc0 + j);
          }
          *(short8*)(As + chunk * 1024 + lane * 16) = *(short8*)va;
        }
        if (sb) {
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) void*)sb,
              (__attribute__((address_space(3))) void*)(Bs + chunk * 1024),
              16, 0, 0);
        } else {
          unsigned short vb[8];
          if constexpr (LB::MAY_SLOW)
            lb.slowload(min(n0 + row, N - 1), kc0, vb);
          else {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              vb[j] = lb.elem(min(n0 + row, N - 1), kc0 + j);
          }
          *(short8*)(Bs + chunk * 1024 + lane * 16) = *(short8*)vb;
        }
      }
    }
  };

  f32x4 acc[4][4] = {};

  if (t_lo >= nk) return;
  // Counted-vmcnt pipeline (guide T3/T4): tile t+1's 8 glds are issued
  // right after the barrier that frees its buffer, and the wait before
  // computing tile t is `vmcnt(8)` — t's own loads retired (in-order),
  // t+1's stay IN FLIGHT across the raw barrier.  __syncthreads() here
  // would emit s_waitcnt vmcnt(0) (glds is a pending LDS write on the
  // VM counter) and drain the prefetch every tile — the measured ~20%
  // stall of the 2-barrier structure.  Straddle/slow chunks take
  // ordinary loads whose count breaks the formula -> those shapes wait
  // vmcnt(0) instead (still pipelined one tile deep).
  const bool counted = !(LA::MAY_SLOW || LB::MAY_SLOW) && (K % 8 == 0);
  stage(0, t_lo);
  int buf = 0;
  for (int t = t_lo; t < nk; ++t) {
    __builtin_amdgcn_s_barrier();      // all waves done reading buf^1
    if (t + 1 < nk) {
      stage(buf ^ 1, t + 1);
      if (counted)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();      // tile t's LDS complete everywhere
    const char* As = smem + buf * NT_LDS_BUF;
    const char* Bs = As + NT_LDS_A;
    const int fr = lane & 15;          // fragment row/col
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int ko = ks * 4 + (lane >> 4);   // k-octet 0..7
      short8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wr * 64 + i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        af[i] = *(const short8*)(As + row * 128 + slot * 16);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wc * 64 + i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        bf[i] = *(const short8*)(Bs + row * 128 + slot * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    buf ^= 1;
  }

  // epilogue: bias + activation + store (row = (l>>4)*4+r, col = l&15);
  // split-K partials accumulate raw fp32 (bias/act applied later)
  const int colw = n0 + wc * 64 + (lane & 15);
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int col = colw + ni * 16;
    if (col >= N) continue;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int rbase = m0 + wr * 64 + mi * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        if (sp > 1) {
          // each K-slice writes its own disjoint fp32 slab (plain
          // stores); the finisher sums the slabs — measured class:
          // slab reduce beats fp32 L2 atomics, and full overwrite
          // removes the zero pass
          ((float*)C)[(long)sid * M * N + (long)row * N + col] =
              acc[mi][ni][r];
          continue;
        }
        // dact != null: this is a dgrad whose consumer's activation
        // derivative is FUSED here (ACT is the derivative code, dact
        // the forward output y) — replaces the separate act_bwd pass
        float v = dact ? acc[mi][ni][r] *
                             act_dy<ACT>(bf2f(dact[(long)row * N + col]))
                       : act_fwd<ACT>(acc[mi][ni][r] + bv);
        if constexpr (sizeof(OUT_T) == 2)
          C[(long)row * N + col] = (OUT_T)f2bf(v);
        else
          C[(long)row * N + col] = (OUT_T)v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// narrow-N NT core (N ≤ 64: CIFAR-class convs with 32/64 kernels, conv
// dgrad into 8..64-channel inputs).  The 128-wide tile computes BN=128
// columns regardless of N, so N=32 wastes 4× of the MFMA work; this
// variant tiles BNT∈{32,64} with the 4 waves stacked 4M×1N (each wave:
// 32 rows × BNT cols), keeping the same glds staging + slot swizzle.
// ---------------------------------------------------------------------------

template <typename LA, typename LB, typename OUT_T, int ACT, int BNT>
__global__ __launch_bounds__(256, 2) void gemm_ntn_k(
    LA la, LB lb, OUT_T* __restrict__ C, const float* __restrict__ bias,
    const unsigned short* __restrict__ dact, int M, int N, int K,
    int tiles_n) {
  constexpr int LDS_B = BNT * BK * 2;
  constexpr int LDS_BUF = NT_LDS_A + LDS_B;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tile = xcd_tile(blockIdx.x, gridDim.x);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * BM, n0 = tn * BNT;

  const int nk = (K + BK - 1) / BK;

  const int rl = lane >> 3;
  const int k8 = lane & 7;
  // loop-invariant per-thread row contexts (see gemm_nt_k)
  typename LA::RowC rca[4];
  typename LB::RowC rcb[BNT / 32];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    rca[i] = la.rowc(min(m0 + (wave * 4 + i) * 8 + rl, M - 1));
#pragma unroll
  for (int i = 0; i < BNT / 32; ++i)
    rcb[i] = lb.rowc(
        min(n0 + (wave * (BNT / 32) + i) * 8 + rl, N - 1));

  auto stage = [&](int buf, int kt) {
    char* As = smem + buf * LDS_BUF;
    char* Bs = As + NT_LDS_A;
    const int k8base = kt * (BK / 8);
    const auto kca0 = la.kc(k8base + (k8 ^ rl));
    const auto kca1 = la.kc(k8base + (k8 ^ rl ^ 1));
    const auto kcb0 = lb.kc(k8base + (k8 ^ rl));
    const auto kcb1 = lb.kc(k8base + (k8 ^ rl ^ 1));
#pragma unroll
    for (int i = 0; i < 4; ++i) {       // A: 16 chunks × 8 rows
      const int chunk = wave * 4 + i;
      const int row = chunk * 8 + rl;
      const int k8e = k8 ^ rl ^ (chunk & 1);
      const int kc0 = (k8base + k8e) * 8;
      const bool straddle = (kc0 < K) & (kc0 + 8 > K);
      const void* sa = straddle
                           ? nullptr
                           : la.addr_rk(rca[i], (chunk & 1) ? kca1 : kca0);
      if (!LA::MAY_SLOW ? !straddle : (sa != nullptr)) {
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)sa,
            (__attribute__((address_space(3))) void*)(As + chunk * 1024),
            16, 0, 0);
      } else {
        unsigned short va[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          va[j] = la.elem(min(m0 + row, M - 1), kc0 + j);
        *(short8*)(As + chunk * 1024 + lane * 16) = *(short8*)va;
      }
    }
#pragma unroll
    for (int i = 0; i < BNT / 32; ++i) {  // B: BNT/8 chunks × 8 rows
      const int chunk = wave * (BNT / 32) + i;
      const int row = chunk * 8 + rl;
      const int k8e = k8 ^ rl ^ (chunk & 1);
      const int kc0 = (k8base + k8e) * 8;
      const bool straddle = (kc0 < K) & (kc0 + 8 > K);
      const void* sb = straddle
                           ? nullptr
                           : lb.addr_rk(rcb[i], (chunk & 1) ? kcb1 : kcb0);
      if (!LB::MAY_SLOW ? !straddle : (sb != nullptr)) {
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)sb,
            (__attribute__((address_space(3))) void*)(Bs + chunk * 1024),
            16, 0, 0);
      } else {
        unsigned short vb[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vb[j] = lb.elem(min(n0 + row, N - 1), kc0 + j);
        *(short8*)(Bs + chunk * 1024 + lane * 16) = *(short8*)vb;
      }
    }
  };

  f32x4 acc[2][BNT / 16] = {};

  // counted-vmcnt pipeline — same discipline as gemm_nt_k above
  const bool counted = !(LA::MAY_SLOW || LB::MAY_SLOW) && (K % 8 == 0);
  constexpr int NLOADS = 4 + BNT / 32;
  stage(0, 0);
  int buf = 0;
  for (int t = 0; t < nk; ++t) {
    __builtin_amdgcn_s_barrier();
    if (t + 1 < nk) {
      stage(buf ^ 1, t + 1);
      if (counted)
        asm volatile("s_waitcnt vmcnt(%0)" :: "i"(NLOADS) : "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    const char* As = smem + buf * LDS_BUF;
    const char* Bs = As + NT_LDS_A;
    const int fr = lane & 15;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int ko = ks * 4 + (lane >> 4);
      short8 af[2], bf[BNT / 16];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int row = wave * 32 + i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        af[i] = *(const short8*)(As + row * 128 + slot * 16);
      }
#pragma unroll
      for (int i = 0; i < BNT / 16; ++i) {
        const int row = i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        bf[i] = *(const short8*)(Bs + row * 128 + slot * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < BNT / 16; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    buf ^= 1;
  }

  const int colw = n0 + (lane & 15);
#pragma unroll
  for (int ni = 0; ni < BNT / 16; ++ni) {
    const int col = colw + ni * 16;
    if (col >= N) continue;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      const int rbase = m0 + wave * 32 + mi * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        float v = dact ? acc[mi][ni][r] *
                             act_dy<ACT>(bf2f(dact[(long)row * N + col]))
                       : act_fwd<ACT>(acc[mi][ni][r] + bv);
        if constexpr (sizeof(OUT_T) == 2)
          C[(long)row * N + col] = (OUT_T)f2bf(v);
        else
          C[(long)row * N + col] = (OUT_T)v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// deep-pipelined narrow-N core: 8 waves (512 thr), BM=256 × BNT=96,
// THREE LDS buffers with a 2-tile prefetch (counted vmcnt(12)) — for
// wait-bound narrow shapes (conv1-class: PMC showed 44% SQ_WAIT at the
// 2-buffer/1-deep schedule; 1 block/CU here, the depth does the hiding).
// LDS per buffer: 32 A-chunks + 16 B-chunk slots (12 real + 4 dummy
// pads staged from the zero page to keep the per-thread glds count
// uniform for the counted wait) = 48 KiB; ×3 = 144 KiB.
// ---------------------------------------------------------------------------
#define N8_CHUNKS 48
#define N8_BUF (N8_CHUNKS * 1024)

template <typename LA, typename LB, typename OUT_T, int ACT,
          int BNT = 96>
__global__ __launch_bounds__(512, 1) void gemm_ntn8_k(
    LA la, LB lb, OUT_T* __restrict__ C, const float* __restrict__ bias,
    const unsigned short* __restrict__ dact, int M, int N, int K,
    int tiles_n) {
  constexpr int BCH = BNT / 8;          // real B chunks
  constexpr int BSTG = BNT > 64 ? 2 : 1;  // B chunks staged per wave
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tile = xcd_tile(blockIdx.x, gridDim.x);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * 256, n0 = tn * BNT;
  const int nk = (K + BK - 1) / BK;
  const int rl = lane >> 3;
  const int k8 = lane & 7;

  typename LA::RowC rca[4];
  typename LB::RowC rcb[BSTG];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    rca[i] = la.rowc(min(m0 + (wave * 4 + i) * 8 + rl, M - 1));
  // B: wave w stages chunk w (+ chunk 8+w when BSTG=2); chunks past
  // BCH are zero-area pads that keep the per-thread glds count uniform
#pragma unroll
  for (int i = 0; i < BSTG; ++i) {
    const int bc = wave + i * 8;
    rcb[i] = lb.rowc(min(n0 + (bc < BCH ? bc * 8 + rl : 0), N - 1));
  }

  auto stage = [&](int buf, int kt) {
    char* base = smem + buf * N8_BUF;
    const int k8base = kt * (BK / 8);
    const auto kca0 = la.kc(k8base + (k8 ^ rl));
    const auto kca1 = la.kc(k8base + (k8 ^ rl ^ 1));
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int chunk = wave * 4 + i;
      const void* sa = la.addr_rk(rca[i], (i & 1) ? kca1 : kca0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)sa,
          (__attribute__((address_space(3))) void*)(base + chunk * 1024),
          16, 0, 0);
    }
    const auto kcb0 = lb.kc(k8base + (k8 ^ rl));
    const auto kcb1 = lb.kc(k8base + (k8 ^ rl ^ 1));
#pragma unroll
    for (int i = 0; i < BSTG; ++i) {
      const int bc = wave + i * 8;
      const int chunk = 32 + bc;
      const void* sb = lb.addr_rk(rcb[i], (bc & 1) ? kcb1 : kcb0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)sb,
          (__attribute__((address_space(3))) void*)(base + chunk * 1024),
          16, 0, 0);
    }
  };

  f32x4 acc[2][BNT / 16] = {};

  if (nk > 0) stage(0, 0);
  if (nk > 1) stage(1, 1);
  int buf = 0;
  for (int t = 0; t < nk; ++t) {
    __builtin_amdgcn_s_barrier();
    constexpr int NL = 4 + BSTG;
    if (t + 2 < nk) {
      stage((buf + 2) % 3, t + 2);
      asm volatile("s_waitcnt vmcnt(%0)" :: "i"(2 * NL) : "memory");
    } else if (t + 1 < nk) {
      asm volatile("s_waitcnt vmcnt(%0)" :: "i"(NL) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    const char* As = smem + buf * N8_BUF;
    const char* Bs = As + 32 * 1024;
    const int fr = lane & 15;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int ko = ks * 4 + (lane >> 4);
      short8 af[2], bf[BNT / 16];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int row = wave * 32 + i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        af[i] = *(const short8*)(As + row * 128 + slot * 16);
      }
#pragma unroll
      for (int i = 0; i < BNT / 16; ++i) {
        const int row = i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        bf[i] = *(const short8*)(Bs + row * 128 + slot * 16);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < BNT / 16; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    buf = buf + 1 == 3 ? 0 : buf + 1;
  }

  const int colw = n0 + (lane & 15);
#pragma unroll
  for (int ni = 0; ni < BNT / 16; ++ni) {
    const int col = colw + ni * 16;
    if (col >= N) continue;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      const int rbase = m0 + wave * 32 + mi * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        float v = dact ? acc[mi][ni][r] *
                             act_dy<ACT>(bf2f(dact[(long)row * N + col]))
                       : act_fwd<ACT>(acc[mi][ni][r] + bv);
        if constexpr (sizeof(OUT_T) == 2)
          C[(long)row * N + col] = (OUT_T)f2bf(v);
        else
          C[(long)row * N + col] = (OUT_T)v;
      }
    }
  }
}

template <typename LA, typename LB, typename OUT_T>
void launch_gemm_ntn8(const LA& la, const LB& lb, OUT_T* C,
                      const float* bias, int M, int N, int K, int act,
                      hipStream_t stream,
                      const unsigned short* dact = nullptr) {
  dim3 block(512);
  if (N <= 32) {
    dim3 grid(((M + 255) / 256) * ((N + 31) / 32));
#define GO8A(A)                                                              \
    hipLaunchKernelGGL((gemm_ntn8_k<LA, LB, OUT_T, A, 32>), grid, block,     \
                       3 * N8_BUF, stream, la, lb, C, bias, dact, M, N,      \
                       K, (N + 31) / 32)
    ACT_SWITCH(act, GO8A);
#undef GO8A
  } else if (N <= 64) {
    dim3 grid(((M + 255) / 256) * ((N + 63) / 64));
#define GO8B(A)                                                              \
    hipLaunchKernelGGL((gemm_ntn8_k<LA, LB, OUT_T, A, 64>), grid, block,     \
                       3 * N8_BUF, stream, la, lb, C, bias, dact, M, N,      \
                       K, (N + 63) / 64)
    ACT_SWITCH(act, GO8B);
#undef GO8B
  } else {
    dim3 grid(((M + 255) / 256) * ((N + 95) / 96));
#define GO8C(A)                                                              \
    hipLaunchKernelGGL((gemm_ntn8_k<LA, LB, OUT_T, A, 96>), grid, block,     \
                       3 * N8_BUF, stream, la, lb, C, bias, dact, M, N,      \
                       K, (N + 95) / 96)
    ACT_SWITCH(act, GO8C);
#undef GO8C
  }
  HIP_CHECK_LAST();
}


// ---------------------------------------------------------------------------
// deep-pipelined WIDE core: 8 waves (512 thr), BM=256 × BN=128, three
// LDS buffers, 2-tile prefetch (counted vmcnt(12)) — the ntn8 schedule
// at full width for the big conv fwd/dgrad shapes.  48 chunks/buffer
// (32 A + 16 B, all real) = 48 KiB; ×3 = 144 KiB, 1 block/CU, 2
// waves/SIMD (same as the 2-block 128² core, but with twice the
// prefetch depth).  Waves stack 8M×1N: each wave owns 32 rows × 128
// cols.
// ---------------------------------------------------------------------------
template <typename LA, typename LB, typename OUT_T, int ACT>
__global__ __launch_bounds__(512, 1) void gemm_nt8d_k(
    LA la, LB lb, OUT_T* __restrict__ C, const float* __restrict__ bias,
    const unsigned short* __restrict__ dact, int M, int N, int K,
    int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tile = xcd_tile(blockIdx.x, gridDim.x);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * 256, n0 = tn * 128;
  const int nk = (K + BK - 1) / BK;
  const int rl = lane >> 3;
  const int k8 = lane & 7;

  typename LA::RowC rca[4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    rca[i] = la.rowc(min(m0 + (wave * 4 + i) * 8 + rl, M - 1));
  typename LB::RowC rcb[2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
    rcb[i] = lb.rowc(min(n0 + (wave + i * 8) * 8 + rl, N - 1));

  auto stage = [&](int buf, int kt) {
    char* base = smem + buf * N8_BUF;
    const int k8base = kt * (BK / 8);
    const auto kca0 = la.kc(k8base + (k8 ^ rl));
    const auto kca1 = la.kc(k8base + (k8 ^ rl ^ 1));
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int chunk = wave * 4 + i;
      const void* sa = la.addr_rk(rca[i], (i & 1) ? kca1 : kca0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)sa,
          (__attribute__((address_space(3))) void*)(base + chunk * 1024),
          16, 0, 0);
    }
    const auto kcb0 = lb.kc(k8base + (k8 ^ rl));
    const auto kcb1 = lb.kc(k8base + (k8 ^ rl ^ 1));
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int bc = wave + i * 8;                 // B chunks 0..15
      const int chunk = 32 + bc;
      const void* sb = lb.addr_rk(rcb[i], (bc & 1) ? kcb1 : kcb0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)sb,
          (__attribute__((address_space(3))) void*)(base + chunk * 1024),
          16, 0, 0);
    }
  };

  f32x4 acc[2][8] = {};

  if (nk > 0) stage(0, 0);
  if (nk > 1) stage(1, 1);
  int buf = 0;
  for (int t = 0; t < nk; ++t) {
    __builtin_amdgcn_s_barrier();
    if (t + 2 < nk) {
      stage(buf + 2 >= 3 ? buf - 1 : buf + 2, t + 2);
      asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
    } else if (t + 1 < nk) {
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    const char* As = smem + buf * N8_BUF;
    const char* Bs = As + 32 * 1024;
    const int fr = lane & 15;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int ko = ks * 4 + (lane >> 4);
      short8 af[2], bf[8];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int row = wave * 32 + i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        af[i] = *(const short8*)(As + row * 128 + slot * 16);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int row = i * 16 + fr;
        const int slot = ko ^ (row & 7) ^ ((row >> 3) & 1);
        bf[i] = *(const short8*)(Bs + row * 128 + slot * 16);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 8; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    buf = buf + 1 == 3 ? 0 : buf + 1;
  }

  const int colw = n0 + (lane & 15);
#pragma unroll
  for (int ni = 0; ni < 8; ++ni) {
    const int col = colw + ni * 16;
    if (col >= N) continue;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      const int rbase = m0 + wave * 32 + mi * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        float v = dact ? acc[mi][ni][r] *
                             act_dy<ACT>(bf2f(dact[(long)row * N + col]))
                       : act_fwd<ACT>(acc[mi][ni][r] + bv);
        if constexpr (sizeof(OUT_T) == 2)
          C[(long)row * N + col] = (OUT_T)f2bf(v);
        else
          C[(long)row * N + col] = (OUT_T)v;
      }
    }
  }
}

template <typename LA, typename LB, typename OUT_T>
void launch_gemm_nt8d(const LA& la, const LB& lb, OUT_T* C,
                      const float* bias, int M, int N, int K, int act,
                      hipStream_t stream,
                      const unsigned short* dact = nullptr) {
  const int tn = (N + 127) / 128;
  dim3 grid(((M + 255) / 256) * tn), block(512);
#define GO8D(A)                                                              \
  hipLaunchKernelGGL((gemm_nt8d_k<LA, LB, OUT_T, A>), grid, block,           \
                     3 * N8_BUF, stream, la, lb, C, bias, dact, M, N, K,     \
                     tn)
  ACT_SWITCH(act, GO8D);
#undef GO8D
  HIP_CHECK_LAST();
}

// split-K finisher: out = act(ws + bias), bf16 or fp32 out
template <typename OUT_T, int ACT>
__global__ void bias_act_cast_k(const float* __restrict__ ws,
                                OUT_T* __restrict__ out,
                                const float* __restrict__ bias,
                                const unsigned short* __restrict__ dact,
                                long n, int N, int nslab) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride) {
    float v = ws[p];
    for (int sl = 1; sl < nslab; ++sl) v += ws[p + (long)sl * n];
    if (dact)
      v = v * act_dy<ACT>(bf2f(dact[p]));
    else
      v = act_fwd<ACT>(v + (bias ? bias[p % N] : 0.f));
    if constexpr (sizeof(OUT_T) == 2)
      out[p] = (OUT_T)f2bf(v);
    else
      out[p] = (OUT_T)v;
  }
}

// ---------------------------------------------------------------------------
// core TN (K-major operands; fp32 out += )
// ---------------------------------------------------------------------------

// dense K-major: element (k, m) at base[k*ld + m]; OOB -> 0.
// loadrow8 fills 8 consecutive k at fixed m — address walks by ld, the
// only per-element work is the bound check (no div/mod in the loop).
struct DenseTN {
  const unsigned short* base;
  long ld;
  int Klim, Mlim;
  __device__ void loadrow8(int k0, int m, unsigned short* v) const {
    const bool mok = m < Mlim;
    long off = (long)k0 * ld + m;
#pragma unroll
    for (int j = 0; j < 8; ++j, off += ld) {
      bool ok = mok & (k0 + j < Klim);
      unsigned short val = base[ok ? off : 0];
      v[j] = ok ? val : (unsigned short)0;
    }
  }
};

// im2col K-major: (k = n*OH*OW + oh*OW + ow, column m = (kh,kw,c)).
// One div/mod pair decodes k0, then (ow, oh, n) walk incrementally —
// the naive per-(k,m) decode was the single hottest ALU path of the
// whole AlexNet step (wgrad staging).
struct Im2colTN {
  const unsigned short* x;
  ConvDesc d;
  int Klim, Mlim;
  __device__ void loadrow8(int k0, int m, unsigned short* v) const {
    unsigned tm = divq((unsigned)m, d.mC);
    int c = m - (int)(tm * d.C);
    unsigned khu = divq(tm, d.mKW);
    int kw = (int)(tm - khu * d.KW), kh = (int)khu;
    const bool mok = (m < Mlim) & (kh < d.KH);
    unsigned tk = divq((unsigned)k0, d.mOW);
    int ow = k0 - (int)(tk * d.OW);
    unsigned nu = divq(tk, d.mOH);
    int oh = (int)(tk - nu * d.OH), n = (int)nu;
    const int iwk = kw - d.PL, ihk = kh - d.PT;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int ih = oh * d.SY + ihk;
      int iw = ow * d.SX + iwk;
      bool ok = mok & (k0 + j < Klim) & ((unsigned)ih < (unsigned)d.H) &
                ((unsigned)iw < (unsigned)d.W);
      long off = ok ? (((long)n * d.H + ih) * d.W + iw) * d.C + c : 0;
      unsigned short val = x[off];
      v[j] = ok ? val : (unsigned short)0;
      if (++ow == d.OW) {
        ow = 0;
        if (++oh == d.OH) { oh = 0; ++n; }
      }
    }
  }
};

// Split-K: gridDim.y slices the K loop; each slice accumulates its
// partial tile with ONE fp32 atomic per element (hw global fadd) — the
// wgrad shapes (M=n_kernels ≤ 384, N=KH·KW·C ≤ a few k, K=N·OH·OW ≈ 10⁵..10⁶)
// have too few MN tiles to fill 256 CUs without it.  beta is applied by
// the launcher (memset / prescale) when split.
template <typename LA, typename LB>
__global__ __launch_bounds__(256, 2) void gemm_tn_k(
    LA la, LB lb, float* __restrict__ C, int M, int N, int K, float beta,
    int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tile = xcd_tile(blockIdx.x, gridDim.x);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN;
  const int wr = wave >> 1, wc = wave & 1;
  const int nk = (K + BK - 1) / BK;
  const int sp = gridDim.y, sid = blockIdx.y;
  const int t_lo = (int)(((long)nk * sid) / sp);
  const int t_hi = (int)(((long)nk * (sid + 1)) / sp);

  // stage one operand's [BK][128] K-major slab as a swizzled [128][BK]
  // image: thread covers (col, k-octet) chunks; loadrow8 walks 8 k's
  // incrementally (coalesced across lanes); one swizzled ds_write_b128.
  auto stage_op = [&](char* dst, const auto& ld_, int kt, int lim0,
                      int is_b) {
    const int k0 = kt * BK;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int chunk = i * 256 + tid;        // 1024 chunks
      const int col = chunk & 127;            // m (or n) within tile
      const int ko = chunk >> 7;              // k-octet 0..7
      unsigned short v[8];
      ld_.loadrow8(k0 + ko * 8, lim0 + col, v);
      const int slot = ko ^ (col & 7);        // bank-conflict-free
      *(short8*)(dst + col * 128 + slot * 16) = *(short8*)v;
    }
    (void)is_b;
  };

  f32x4 acc[4][4] = {};
  char* As0 = smem;
  char* Bs0 = smem + NT_LDS_A;

  for (int t = t_lo; t < t_hi; ++t) {
    const int buf = t & 1;
    char* As = As0 + buf * NT_LDS_BUF;
    char* Bs = Bs0 + buf * NT_LDS_BUF;
    stage_op(As, la, t, m0, 0);
    stage_op(Bs, lb, t, n0, 1);
    __syncthreads();
    const int fr = lane & 15;
    const int fo = lane >> 4;                 // k-octet 0..3 (+4 for ks=1)
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      short8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wr * 64 + i * 16 + fr;
        const int slot = (ks * 4 + fo) ^ (row & 7);
        af[i] = *(const short8*)(As + row * 128 + slot * 16);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wc * 64 + i * 16 + fr;
        const int slot = (ks * 4 + fo) ^ (row & 7);
        bf[i] = *(const short8*)(Bs + row * 128 + slot * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  const int colw = n0 + wc * 64 + (lane & 15);
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int col = colw + ni * 16;
    if (col >= N) continue;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int rbase = m0 + wr * 64 + mi * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        float* p = C + (long)row * N + col;
        if (sp > 1)
          unsafeAtomicAdd(p, acc[mi][ni][r]);
        else
          *p = (beta == 0.f ? 0.f : beta * *p) + acc[mi][ni][r];
      }
    }
  }
}

__global__ void scale_f32_k(float* __restrict__ p, long n, float beta) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long q = i; q < n; q += stride) p[q] *= beta;
}

// ---------------------------------------------------------------------------
// core TN v2 — m-contiguous 16-B staged loads (the v1 per-element
// k-gather made hipcc branch around each 2-B load and wait vmcnt(0)
// per element — the documented conditional-load trap; one select per
// 16-B chunk here).  LDS image [BK][BM+2] u16 (row pad ⇒ the four
// 16-lane k-octet groups of a fragment read hit distinct bank halves);
// fragments assembled from 8 ds_read_u16.  Requires both operand
// m-dims % 8 == 0 (true for every conv/FC shape in the framework;
// launcher falls back to v1 otherwise).
// ---------------------------------------------------------------------------

#define TNP (BM + 16)                   // padded LDS row: stride ≡ 8
                                        // dwords (mod 64) ⇒ the 4 rows
                                        // of a tr_b16 block hit
                                        // disjoint bank octets
#define TBK 32                          // tn2 K-step: half of BK so the
                                        // LDS fits 3-4 blocks/CU and the
                                        // stage latency hides behind
                                        // more resident waves
#define TN2_LDS_OP (TBK * TNP * 2)      // one operand image, bytes
#define TN2_LDS_BUF (TN2_LDS_OP * 2)    // A + B
#define TN2_LDS_TOTAL (TN2_LDS_BUF * 2) // double-buffered

struct DenseTNv {
  static constexpr bool HAS_ADDR16 = true;
  const unsigned short* base;
  long ld;
  int Klim, Mlim;
  // T14 split API: address+bounds now, raw load by the caller, the
  // ok?v:0 select deferred to the LDS-write pass (a select at load
  // time forces the vmcnt wait at ISSUE, defeating the async split)
  __device__ const unsigned short* addr16(int k, int mo, bool& ok) const {
    ok = (k < Klim) & (mo * 8 + 8 <= Mlim);
    return ok ? base + (long)k * ld + mo * 8 : base;
  }
  __device__ void load16(int k, int mo, unsigned short* v) const {
    bool ok;
    const unsigned short* p = addr16(k, mo, ok);
    *(short8*)v = *(const short8*)p;     // unconditional 16-B load
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = ok ? v[j] : (unsigned short)0;
  }
};

struct Im2colTNv {
  static constexpr bool HAS_ADDR16 = true;
  const unsigned short* x;
  ConvDesc d;
  int Klim, Mlim;
  __device__ const unsigned short* addr16(int k, int mo, bool& ok) const {
    const int m = mo * 8;
    unsigned tm = divq((unsigned)m, d.mC);   // c stays in one C-run
    int c = m - (int)(tm * d.C);
    unsigned khu = divq(tm, d.mKW);
    int kw = (int)(tm - khu * d.KW), kh = (int)khu;
    unsigned tk = divq((unsigned)k, d.mOW);
    int ow = k - (int)(tk * d.OW);
    unsigned nu = divq(tk, d.mOH);
    int oh = (int)(tk - nu * d.OH), n = (int)nu;
    int ih = oh * d.SY - d.PT + kh;
    int iw = ow * d.SX - d.PL + kw;
    ok = (k < Klim) & (m < Mlim) & (kh < d.KH) &
         ((unsigned)ih < (unsigned)d.H) &
         ((unsigned)iw < (unsigned)d.W);
    return ok ? x + (((long)n * d.H + ih) * d.W + iw) * d.C + c : x;
  }
  __device__ void load16(int k, int mo, unsigned short* v) const {
    bool ok;
    const unsigned short* p = addr16(k, mo, ok);
    *(short8*)v = *(const short8*)p;
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = ok ? v[j] : (unsigned short)0;
  }
};

// packed C%4 variant (RGB padded to 4): a SEPARATE type so the
// fallback machinery costs the C%8 wgrad instantiations nothing
struct Im2colTNvP {
  static constexpr bool HAS_ADDR16 = false;  // refused-chunk machinery
  const unsigned short* x;
  ConvDesc d;
  int Klim, Mlim;
  __device__ void load16(int k, int mo, unsigned short* v) const {
    const int m = mo * 8;
    unsigned tm = divq((unsigned)m, d.mC);
    int c = m - (int)(tm * d.C);
    unsigned khu = divq(tm, d.mKW);
    int kwm = (int)(tm - khu * d.KW), kh = (int)khu;
    unsigned tk = divq((unsigned)k, d.mOW);
    int ow = k - (int)(tk * d.OW);
    unsigned nu = divq(tk, d.mOH);
    int oh = (int)(tk - nu * d.OH), n = (int)nu;
    int ih = oh * d.SY - d.PT + kh;
    int iw = ow * d.SX - d.PL + kwm;
    const int rowlen = d.KW * d.C;
    long off = (((long)n * d.H + ih) * d.W + iw) * d.C + c;
    const bool fast = ((m % rowlen) + 8 <= rowlen) &&
                      !(d.PT | d.PL) && !(off & 3) && (k < Klim) &&
                      (m + 8 <= Mlim) && (kh < d.KH) &&
                      ((unsigned)ih < (unsigned)d.H);
    if (fast) {
      // two 8-B loads: off is only guaranteed 4-elem aligned
      *(unsigned long long*)v = *(const unsigned long long*)(x + off);
      *((unsigned long long*)v + 1) =
          *((const unsigned long long*)(x + off) + 1);
      return;
    }
    const int pos = m % rowlen;
    if (!(d.PT | d.PL) && rowlen - pos == 4 && !(d.C & 3) &&
        (k < Klim) && (m + 8 <= Mlim)) {
      // clean kh crossing: two contiguous 8-B-aligned runs (4+4)
      unsigned long long r1 = 0, r2 = 0;
      if (kh < d.KH && (unsigned)ih < (unsigned)d.H)
        r1 = *(const unsigned long long*)(x + off);
      if (kh + 1 < d.KH && (unsigned)(ih + 1) < (unsigned)d.H)
        r2 = *(const unsigned long long*)(
            x + (((long)n * d.H + ih + 1) * d.W + ow * d.SX - d.PL) *
                    d.C);
      *(unsigned long long*)v = r1;
      *((unsigned long long*)v + 1) = r2;
      return;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int mj = m + j;
      int cj = mj % d.C, tj = mj / d.C;
      int kwj = tj % d.KW, khj = tj / d.KW;
      int ihj = oh * d.SY - d.PT + khj;
      int iwj = ow * d.SX - d.PL + kwj;
      const bool okj = (k < Klim) & (mj < Mlim) & (khj < d.KH) &
                       ((unsigned)ihj < (unsigned)d.H) &
                       ((unsigned)iwj < (unsigned)d.W);
      v[j] = okj ? x[(((long)n * d.H + ihj) * d.W + iwj) * d.C + cj]
                 : (unsigned short)0;
    }
  }
};

template <typename LA, typename LB>
__global__ __launch_bounds__(256, 2) void gemm_tn2_k(
    LA la, LB lb, float* __restrict__ C, int M, int N, int K, float beta,
    int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tile = xcd_tile(blockIdx.x, gridDim.x);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN;
  const int wr = wave >> 1, wc = wave & 1;
  const int nk = (K + TBK - 1) / TBK;
  const int sp = gridDim.y, sid = blockIdx.y;
  const int t_lo = (int)(((long)nk * sid) / sp);
  const int t_hi = (int)(((long)nk * (sid + 1)) / sp);
  const int moA = m0 >> 3, moB = n0 >> 3;

  auto stage2 = [&](int buf, int kt) {
    char* As = smem + buf * TN2_LDS_BUF;
    char* Bs = As + TN2_LDS_OP;
    const int k0t = kt * TBK;
    const int oct = tid & 15;
    const int kk0 = tid >> 4;            // 16 k-rows per pass
#pragma unroll
    for (int pass = 0; pass < TBK / 16; ++pass) {
      const int kk = pass * 16 + kk0;
      unsigned short v[8];
      la.load16(k0t + kk, moA + oct, v);
      *(short8*)(As + (kk * TNP + oct * 8) * 2) = *(short8*)v;
      lb.load16(k0t + kk, moB + oct, v);
      *(short8*)(Bs + (kk * TNP + oct * 8) * 2) = *(short8*)v;
    }
  };

  f32x4 acc[4][4] = {};
  if (t_lo >= t_hi) return;

  // T14 async-stage split (PMC round 2: waves were 48% parked in
  // waits, MFMA 26% busy, with the synchronous load->ds_write stage):
  // tile t+1's raw 16-B loads are ISSUED one full compute phase before
  // their vmcnt wait; the bounds select + ds_write happen after the
  // barrier that frees the target buffer, and the next tile's loads
  // re-issue immediately so they span the following compute.
  unsigned short vA[TBK / 16][8], vB[TBK / 16][8];
  bool okA[TBK / 16], okB[TBK / 16];
  auto load_regs = [&](int kt) {
    if constexpr (LA::HAS_ADDR16 && LB::HAS_ADDR16) {
      const int k0t = kt * TBK;
      const int oct = tid & 15;
      const int kk0 = tid >> 4;
#pragma unroll
      for (int pass = 0; pass < TBK / 16; ++pass) {
        const int kk = pass * 16 + kk0;
        *(short8*)vA[pass] =
            *(const short8*)la.addr16(k0t + kk, moA + oct, okA[pass]);
        *(short8*)vB[pass] =
            *(const short8*)lb.addr16(k0t + kk, moB + oct, okB[pass]);
      }
    }
  };
  auto write_lds = [&](int buf_) {
    char* As_ = smem + buf_ * TN2_LDS_BUF;
    char* Bs_ = As_ + TN2_LDS_OP;
    const int oct = tid & 15;
    const int kk0 = tid >> 4;
#pragma unroll
    for (int pass = 0; pass < TBK / 16; ++pass) {
      const int kk = pass * 16 + kk0;
      short8 a = *(short8*)vA[pass], b = *(short8*)vB[pass];
      *(short8*)(As_ + (kk * TNP + oct * 8) * 2) =
          okA[pass] ? a : (short8)0;
      *(short8*)(Bs_ + (kk * TNP + oct * 8) * 2) =
          okB[pass] ? b : (short8)0;
    }
  };

  int buf = 0;
  if constexpr (LA::HAS_ADDR16 && LB::HAS_ADDR16) {
    load_regs(t_lo);
    write_lds(0);
    if (t_lo + 1 < t_hi) load_regs(t_lo + 1);
    __syncthreads();
  } else {
    stage2(0, t_lo);
    __syncthreads();
  }
  for (int t = t_lo; t < t_hi; ++t) {
    if constexpr (!(LA::HAS_ADDR16 && LB::HAS_ADDR16)) {
      if (t + 1 < t_hi) stage2(buf ^ 1, t + 1);
    }
    const unsigned short* As =
        (const unsigned short*)(smem + buf * TN2_LDS_BUF);
    const unsigned short* Bs = As + TBK * TNP;
    // hardware-transpose fragment reads (gfx950 ds_read_b64_tr_b16 via
    // the compiler-tracked builtin — raw asm with truncated generic
    // pointers + hand waitcnt raced non-deterministically under
    // multi-block residency): lanes 4a..4a+3 of each 16-lane group
    // fetch the 4 8-B chunks of image row kb+a cols [m0f, m0f+16); the
    // group transpose hands lane j its m-column across the 4 k's — 2
    // reads per fragment vs 8 scalar u16 (probe: tools/probe_trb16.hip)
    typedef __attribute__((ext_vector_type(4))) short v4s;
    typedef __attribute__((address_space(3))) v4s* lds_v4s_ptr;
    const int gl = lane & 15;
    const int a4 = gl >> 2;              // k-row within the quad
    const int c4 = gl & 3;               // 8-B chunk within the row
    const int ko8 = (lane >> 4) * 8;     // k-octet base within 32
#pragma unroll
    for (int ks = 0; ks < TBK / 32; ++ks) {
      const int kb = ks * 32 + ko8;
      const int rowe = (kb + a4) * TNP + c4 * 4;
      short8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const unsigned short* ap = As + rowe + (wr * 64 + i * 16);
        v4s lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4s_ptr)ap);
        v4s hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4s_ptr)(ap + 4 * TNP));
        *(v4s*)&af[i] = lo;
        *((v4s*)&af[i] + 1) = hi;
        const unsigned short* bp = Bs + rowe + (wc * 64 + i * 16);
        lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_v4s_ptr)bp);
        hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4s_ptr)(bp + 4 * TNP));
        *(v4s*)&bf[i] = lo;
        *((v4s*)&bf[i] + 1) = hi;
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    if constexpr (LA::HAS_ADDR16 && LB::HAS_ADDR16) {
      // write target buf^1 was freed by the PREVIOUS iteration's
      // trailing barrier (tile t-1's reads all precede it), and
      // compute(t) above reads buf — no conflict, so one barrier per
      // tile: writes visible before the next iteration's reads.
      if (t + 1 < t_hi) {
        write_lds(buf ^ 1);            // tile t+1 from the in-flight regs
        if (t + 2 < t_hi) load_regs(t + 2);  // re-issue immediately
      }
    }
    __syncthreads();
    buf ^= 1;
  }

  const int colw = n0 + wc * 64 + (lane & 15);
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int col = colw + ni * 16;
    if (col >= N) continue;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int rbase = m0 + wr * 64 + mi * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        float* p = C + (long)row * N + col;
        if (sp > 1)
          unsafeAtomicAdd(p, acc[mi][ni][r]);
        else
          *p = (beta == 0.f ? 0.f : beta * *p) + acc[mi][ni][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// narrow-M TN2 core (M ≤ 64: wgrad of 32/64-kernel conv layers, where
// M = n_kernels and the 128-row tile wastes 2-4× of the MFMA work).
// BMT∈{32,64} M-rows per tile, 4 waves stacked 1M×4N (each wave: all
// BMT rows × 32 n-cols).  The A LDS image keeps the TNP row pitch —
// tr_b16 requires pitch ≡ 8 dwords (mod 64) — so only compute shrinks,
// not staging legality.  Split-K atomics as in gemm_tn2_k.
// ---------------------------------------------------------------------------

template <typename LA, typename LB, int BMT>
__global__ __launch_bounds__(256, 2) void gemm_tn2n_k(
    LA la, LB lb, float* __restrict__ C, int M, int N, int K, float beta,
    int tiles_n, float* __restrict__ bias_out) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tile = xcd_tile(blockIdx.x, gridDim.x);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * BMT, n0 = tn * BN;
  const int nk = (K + TBK - 1) / TBK;
  const int sp = gridDim.y, sid = blockIdx.y;
  const int t_lo = (int)(((long)nk * sid) / sp);
  const int t_hi = (int)(((long)nk * (sid + 1)) / sp);
  const int moA = m0 >> 3, moB = n0 >> 3;

  // fused bias gradient: A IS the (activation-folded) err_output, so
  // each thread co-accumulates the column sums of the A elements it
  // stages — replaces the separate col_sum kernel pair for this layer.
  // Only tn==0 blocks contribute (A is identical across n-tiles).
  float bsum[8] = {};
  const bool do_bias = bias_out != nullptr && tn == 0;

  auto stage2 = [&](int buf, int kt) {
    char* As = smem + buf * TN2_LDS_BUF;
    char* Bs = As + TN2_LDS_OP;
    const int k0t = kt * TBK;
    {
      // A: BMT/8 octets × TBK k-rows
      const int oa = tid & (BMT / 8 - 1);
      const int ka = tid / (BMT / 8);
      if (ka < TBK) {
        unsigned short v[8];
        la.load16(k0t + ka, moA + oa, v);
        *(short8*)(As + (ka * TNP + oa * 8) * 2) = *(short8*)v;
        if (do_bias)
#pragma unroll
          for (int j = 0; j < 8; ++j) bsum[j] += bf2f(v[j]);
      }
    }
    const int oct = tid & 15;
    const int kk0 = tid >> 4;
#pragma unroll
    for (int pass = 0; pass < TBK / 16; ++pass) {
      const int kk = pass * 16 + kk0;
      unsigned short v[8];
      lb.load16(k0t + kk, moB + oct, v);
      *(short8*)(Bs + (kk * TNP + oct * 8) * 2) = *(short8*)v;
    }
  };

  f32x4 acc[BMT / 16][2] = {};
  if (t_lo >= t_hi) return;

  // T14 async-stage split (see gemm_tn2_k): raw loads issue a full
  // compute phase ahead; bounds select, bias co-accumulation and
  // ds_write happen in the write pass after the barrier.
  unsigned short nvA[8], nvB[TBK / 16][8];
  bool nokA = false, nokB[TBK / 16];
  const int oaA = tid & (BMT / 8 - 1);
  const int kaA = tid / (BMT / 8);
  auto load_regs = [&](int kt) {
    if constexpr (LA::HAS_ADDR16 && LB::HAS_ADDR16) {
      const int k0t = kt * TBK;
      if (kaA < TBK)
        *(short8*)nvA =
            *(const short8*)la.addr16(k0t + kaA, moA + oaA, nokA);
      const int oct = tid & 15;
      const int kk0 = tid >> 4;
#pragma unroll
      for (int pass = 0; pass < TBK / 16; ++pass) {
        const int kk = pass * 16 + kk0;
        *(short8*)nvB[pass] =
            *(const short8*)lb.addr16(k0t + kk, moB + oct, nokB[pass]);
      }
    }
  };
  auto write_lds = [&](int buf_) {
    char* As_ = smem + buf_ * TN2_LDS_BUF;
    char* Bs_ = As_ + TN2_LDS_OP;
    if (kaA < TBK) {
      short8 a = *(short8*)nvA;
      a = nokA ? a : (short8)0;
      *(short8*)(As_ + (kaA * TNP + oaA * 8) * 2) = a;
      if (do_bias)
#pragma unroll
        for (int j = 0; j < 8; ++j) bsum[j] += bf2f(((unsigned short*)&a)[j]);
    }
    const int oct = tid & 15;
    const int kk0 = tid >> 4;
#pragma unroll
    for (int pass = 0; pass < TBK / 16; ++pass) {
      const int kk = pass * 16 + kk0;
      short8 b = *(short8*)nvB[pass];
      *(short8*)(Bs_ + (kk * TNP + oct * 8) * 2) =
          nokB[pass] ? b : (short8)0;
    }
  };

  int buf = 0;
  if constexpr (LA::HAS_ADDR16 && LB::HAS_ADDR16) {
    load_regs(t_lo);
    write_lds(0);
    if (t_lo + 1 < t_hi) load_regs(t_lo + 1);
    __syncthreads();
  } else {
    stage2(0, t_lo);
    __syncthreads();
  }
  for (int t = t_lo; t < t_hi; ++t) {
    if constexpr (!(LA::HAS_ADDR16 && LB::HAS_ADDR16)) {
      if (t + 1 < t_hi) stage2(buf ^ 1, t + 1);
    }
    const unsigned short* As =
        (const unsigned short*)(smem + buf * TN2_LDS_BUF);
    const unsigned short* Bs = As + TBK * TNP;
    typedef __attribute__((ext_vector_type(4))) short v4s;
    typedef __attribute__((address_space(3))) v4s* lds_v4s_ptr;
    const int gl = lane & 15;
    const int a4 = gl >> 2;
    const int c4 = gl & 3;
    const int ko8 = (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < TBK / 32; ++ks) {
      const int kb = ks * 32 + ko8;
      const int rowe = (kb + a4) * TNP + c4 * 4;
      short8 af[BMT / 16], bf[2];
#pragma unroll
      for (int i = 0; i < BMT / 16; ++i) {
        const unsigned short* ap = As + rowe + i * 16;
        v4s lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4s_ptr)ap);
        v4s hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4s_ptr)(ap + 4 * TNP));
        *(v4s*)&af[i] = lo;
        *((v4s*)&af[i] + 1) = hi;
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const unsigned short* bp = Bs + rowe + (wave * 32 + j * 16);
        v4s lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4s_ptr)bp);
        v4s hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
            (lds_v4s_ptr)(bp + 4 * TNP));
        *(v4s*)&bf[j] = lo;
        *((v4s*)&bf[j] + 1) = hi;
      }
#pragma unroll
      for (int mi = 0; mi < BMT / 16; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    if constexpr (LA::HAS_ADDR16 && LB::HAS_ADDR16) {
      if (t + 1 < t_hi) {
        write_lds(buf ^ 1);
        if (t + 2 < t_hi) load_regs(t + 2);
      }
    }
    __syncthreads();
    buf ^= 1;
  }

  if (do_bias) {
    // cross-thread reduce of the staged-A column sums (LDS is free
    // after the loop's trailing barrier), one atomic per m-column
    float* red = (float*)smem;
    const int oa = tid & (BMT / 8 - 1);
    const int ka = tid / (BMT / 8);
    if (ka < TBK) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        red[ka * BMT + oa * 8 + j] = bsum[j];
    }
    __syncthreads();
    if (tid < BMT && m0 + tid < M) {
      float s = 0.f;
#pragma unroll 4
      for (int r = 0; r < TBK; ++r) s += red[r * BMT + tid];
      if (sp > 1)
        unsafeAtomicAdd(bias_out + m0 + tid, s);
      else
        bias_out[m0 + tid] =
            (beta == 0.f ? 0.f : beta * bias_out[m0 + tid]) + s;
    }
  }

  const int colw = n0 + wave * 32 + (lane & 15);
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int col = colw + ni * 16;
    if (col >= N) continue;
#pragma unroll
    for (int mi = 0; mi < BMT / 16; ++mi) {
      const int rbase = m0 + mi * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rbase + r;
        if (row >= M) continue;
        float* p = C + (long)row * N + col;
        if (sp > 1)
          unsafeAtomicAdd(p, acc[mi][ni][r]);
        else
          *p = (beta == 0.f ? 0.f : beta * *p) + acc[mi][ni][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launch helpers (called from bindings.cpp)
// ---------------------------------------------------------------------------

static inline int cdiv(int a, int b) { return (a + b - 1) / b; }

// ---------------------------------------------------------------------------
// tunable routing table (SURVEY §2b autotune-DB row): defaults are the
// measured round-1/2 heuristics; veles.znicz_amd.ops loads the swept
// gfx950_routing.json at import and overrides them via set_routing()
// (regenerate with tools/gen_routing.py on an MI355X).
// ---------------------------------------------------------------------------
int g_nt_split_budget = 512;    // FC NT split-K: blocks to fill toward
int g_tn_split_budget = 1024;   // TN wgrad split-K budget
int g_tn2_split_budget = 1024;  // TN2 wgrad split-K budget
int g_ntn_max_n = 96;           // widest narrow-N NT tile routed

void set_routing(int nt_split, int tn_split, int tn2_split,
                 int ntn_max_n) {
  if (nt_split > 0) g_nt_split_budget = nt_split;
  if (tn_split > 0) g_tn_split_budget = tn_split;
  if (tn2_split > 0) g_tn2_split_budget = tn2_split;
  if (ntn_max_n >= 0) g_ntn_max_n = ntn_max_n;
}

// Per-shape split table for the DENSE (FC) entry points only — the
// device_infos.json idea one level deeper: exact (M,N,K) rows swept on
// hardware (tools/gen_fc_splits.py) override the budget formula, which
// over-splits small-tile FC shapes (measured: fc8 wgrad 33.1→12.9 µs,
// fc8 fwd 26.0→19.0 µs at batch 256).  Conv GEMMs never consult this.
// Populated at import / tool time, read-only on the hot path.
static std::unordered_map<unsigned long long, int> g_fc_nt_splits;
static std::unordered_map<unsigned long long, int> g_fc_tn_splits;

static inline unsigned long long fc_shape_key(int M, int N, int K) {
  return ((unsigned long long)(unsigned)M << 42) |
         ((unsigned long long)(unsigned)N << 21) |
         (unsigned long long)(unsigned)K;
}

void set_fc_split(int is_tn, int M, int N, int K, int splits) {
  auto& t = is_tn ? g_fc_tn_splits : g_fc_nt_splits;
  if (splits <= 0)
    t.erase(fc_shape_key(M, N, K));
  else
    t[fc_shape_key(M, N, K)] = splits;
}

static inline int fc_split_lookup(bool is_tn, int M, int N, int K) {
  const auto& t = is_tn ? g_fc_tn_splits : g_fc_nt_splits;
  auto it = t.find(fc_shape_key(M, N, K));
  return it == t.end() ? 0 : it->second;   // 0 = use the budget formula
}


template <typename LA, typename LB, typename OUT_T>
void launch_gemm_nt(const LA& la, const LB& lb, OUT_T* C, const float* bias,
                    int M, int N, int K, int act, hipStream_t stream,
                    const unsigned short* dact = nullptr) {
  if (N <= 32) {
    const int tn = cdiv(N, 32);
    dim3 grid(cdiv(M, BM) * tn), block(256);
#define GO32(A)                                                              \
    hipLaunchKernelGGL((gemm_ntn_k<LA, LB, OUT_T, A, 32>), grid, block,      \
                       (NT_LDS_A + 32 * BK * 2) * 2, stream, la, lb, C,      \
                       bias, dact, M, N, K, tn)
    ACT_SWITCH(act, GO32);
#undef GO32
    HIP_CHECK_LAST();
    return;
  }
  if (N <= 64) {
    const int tn = cdiv(N, 64);
    dim3 grid(cdiv(M, BM) * tn), block(256);
#define GO64(A)                                                              \
    hipLaunchKernelGGL((gemm_ntn_k<LA, LB, OUT_T, A, 64>), grid, block,      \
                       (NT_LDS_A + 64 * BK * 2) * 2, stream, la, lb, C,      \
                       bias, dact, M, N, K, tn)
    ACT_SWITCH(act, GO64);
#undef GO64
    HIP_CHECK_LAST();
    return;
  }
  if (N <= 96 && g_ntn_max_n >= 96 && LA::N96_OK) {
    // AlexNet conv1 class (K=96 kernels): the 128-wide tile would
    // waste 25% of the MFMA columns
    const int tn = cdiv(N, 96);
    dim3 grid(cdiv(M, BM) * tn), block(256);
#define GO96(A)                                                              \
    hipLaunchKernelGGL((gemm_ntn_k<LA, LB, OUT_T, A, 96>), grid, block,      \
                       (NT_LDS_A + 96 * BK * 2) * 2, stream, la, lb, C,      \
                       bias, dact, M, N, K, tn)
    ACT_SWITCH(act, GO96);
#undef GO96
    HIP_CHECK_LAST();
    return;
  }
  const int tn = cdiv(N, BN);
  dim3 grid(cdiv(M, BM) * tn), block(256);
#define GO(A)                                                                \
  hipLaunchKernelGGL((gemm_nt_k<LA, LB, OUT_T, A>), grid, block,             \
                     NT_LDS_TOTAL, stream, la, lb, C, bias, dact, M, N, K, \
                     tn)
  ACT_SWITCH(act, GO);
#undef GO
  HIP_CHECK_LAST();
}

template <typename LA, typename LB>
void launch_gemm_tn(const LA& la, const LB& lb, float* C, int M, int N,
                    int K, float beta, hipStream_t stream) {
  const int tn = cdiv(N, BN);
  const int tiles = cdiv(M, BM) * tn;
  const int nk = cdiv(K, BK);
  const int splits = max(1, min(nk, g_tn_split_budget / tiles));
  if (splits > 1) {
    if (beta == 0.f) {
      void zero_f32_launch(float*, long, hipStream_t);
      zero_f32_launch(C, (long)M * N, stream);
    } else if (beta != 1.f) {
      long n = (long)M * N;
      int b = (int)min((n + 255) / 256, (long)2048);
      hipLaunchKernelGGL(scale_f32_k, dim3(b), dim3(256), 0, stream, C, n,
                         beta);
    }
  }
  dim3 grid(tiles, splits), block(256);
  hipLaunchKernelGGL((gemm_tn_k<LA, LB>), grid, block, NT_LDS_TOTAL, stream,
                     la, lb, C, M, N, K, beta, tn);
  HIP_CHECK_LAST();
}

template <typename LA, typename LB>
bool launch_gemm_tn2(const LA& la, const LB& lb, float* C, int M, int N,
                     int K, float beta, hipStream_t stream,
                     float* bias_out = nullptr, int split_override = 0) {
  const int tn = cdiv(N, BN);
  const int bmt = M <= 32 ? 32 : (M <= 64 ? 64 : BM);
  // bias fusion is narrow-tile only: in the 128-row kernel the extra
  // staging registers cost more on the big fc/conv wgrads than the
  // separate col_sum kernel does
  if (bmt == BM) bias_out = nullptr;
  const int tiles = cdiv(M, bmt) * tn;
  const int nk = cdiv(K, TBK);
  // each split must carry >= 4 K-tiles: tiny-K wgrads (K = batch,
  // nk ~ 8) otherwise split into atomic-heavy slivers (fc8 wgrad
  // measured 23 TF/s with 2-tile splits)
  int splits =
      max(1, min(min(nk, g_tn2_split_budget / tiles), max(1, nk / 4)));
  if (split_override > 0) splits = min(nk, split_override);
  if (splits > 1) {
    if (beta == 0.f) {
      void zero_f32_launch(float*, long, hipStream_t);
      void zero2_f32_launch(float*, long, float*, long, hipStream_t);
      if (bias_out)
        zero2_f32_launch(C, (long)M * N, bias_out, M, stream);
      else
        zero_f32_launch(C, (long)M * N, stream);
    } else if (beta != 1.f) {
      long n = (long)M * N;
      int b = (int)min((n + 255) / 256, (long)2048);
      hipLaunchKernelGGL(scale_f32_k, dim3(b), dim3(256), 0, stream, C, n,
                         beta);
      if (bias_out)
        hipLaunchKernelGGL(scale_f32_k, dim3(1), dim3(256), 0, stream,
                           bias_out, (long)M, beta);
    }
  }
  dim3 grid(tiles, splits), block(256);
  if (bmt == 32)
    hipLaunchKernelGGL((gemm_tn2n_k<LA, LB, 32>), grid, block,
                       TN2_LDS_TOTAL, stream, la, lb, C, M, N, K, beta,
                       tn, bias_out);
  else if (bmt == 64)
    hipLaunchKernelGGL((gemm_tn2n_k<LA, LB, 64>), grid, block,
                       TN2_LDS_TOTAL, stream, la, lb, C, M, N, K, beta,
                       tn, bias_out);
  else
    hipLaunchKernelGGL((gemm_tn2_k<LA, LB>), grid, block, TN2_LDS_TOTAL,
                       stream, la, lb, C, M, N, K, beta, tn);
  HIP_CHECK_LAST();
  return bias_out != nullptr;
}

// explicit entry points used by bindings.cpp --------------------------------

void gemm_nt_dense(const unsigned short* A, const unsigned short* B,
                   void* C, bool c_is_f32, const float* bias,
                   const unsigned short* zero, int M, int N, int K,
                   int act, hipStream_t s,
                   const unsigned short* dact = nullptr) {
  DenseNT la{A, zero, K, K}, lb{B, zero, K, K};
  const int tiles = cdiv(M, BM) * cdiv(N, BN);
  const int nk = cdiv(K, BK);
  int splits = max(1, min(nk, g_nt_split_budget / tiles));
  if (int ov = fc_split_lookup(false, M, N, K))
    splits = max(1, min(nk, ov));
  if (splits > 1) {
    // fp32 workspace + atomic partials, then one bias/act/cast pass.
    // ONE process-global workspace: every FC split-K launch in this
    // engine runs on the compute stream (or its hipGraph capture
    // proxy), strictly serialized by the unit graph — the comm stream
    // never calls gemm_nt_dense.  A per-stream map is NOT usable here:
    // torch captures on a fresh internal stream, and a lazy hipMalloc
    // during capture invalidates the whole hipGraph.  Warmup steps
    // (always before capture) size the buffer for the model's shapes.
    static std::mutex ws_mu;
    static float* ws = nullptr;
    static size_t cap = 0;
    size_t need = (size_t)splits * M * N * sizeof(float);
    {
      std::lock_guard<std::mutex> lk(ws_mu);
      if (need > cap) {
        if (ws) (void)hipFree(ws);
        cap = need < (32u << 20) ? (32u << 20) : need;
        if (hipMalloc(&ws, cap) != hipSuccess) {
          ws = nullptr;
          cap = 0;
          throw std::runtime_error("gemm_nt split workspace alloc");
        }
      }
    }
    dim3 grid(tiles, splits), block(256);
    hipLaunchKernelGGL((gemm_nt_k<DenseNT, DenseNT, float, ACT_LINEAR>),
                       grid, block, NT_LDS_TOTAL, s, la, lb, ws,
                       (const float*)nullptr,
                       (const unsigned short*)nullptr, M, N, K,
                       cdiv(N, BN));
    HIP_CHECK_LAST();
    long n = (long)M * N;
    int b = (int)min((n + 255) / 256, (long)2048);
#define FIN(A_)                                                              \
    do {                                                                     \
      if (c_is_f32)                                                          \
        hipLaunchKernelGGL((bias_act_cast_k<float, A_>), dim3(b),            \
                           dim3(256), 0, s, ws, (float*)C, bias, dact, n,    \
                           N, splits);                                       \
      else                                                                   \
        hipLaunchKernelGGL((bias_act_cast_k<unsigned short, A_>), dim3(b),   \
                           dim3(256), 0, s, ws, (unsigned short*)C, bias,    \
                           dact, n, N, splits);                              \
    } while (0)
    ACT_SWITCH(act, FIN);
#undef FIN
    HIP_CHECK_LAST();
    return;
  }
  if (c_is_f32)
    launch_gemm_nt(la, lb, (float*)C, bias, M, N, K, act, s, dact);
  else
    launch_gemm_nt(la, lb, (unsigned short*)C, bias, M, N, K, act, s,
                   dact);
}

bool gemm_tn_dense(const unsigned short* A, const unsigned short* B,
                   float* C, int M, int N, int K, float beta,
                   hipStream_t s, float* bias_out) {
  if ((M & 7) == 0 && (N & 7) == 0) {
    DenseTNv la{A, M, K, M}, lb{B, N, K, N};
    return launch_gemm_tn2(la, lb, C, M, N, K, beta, s, bias_out,
                           fc_split_lookup(true, M, N, K));
  }
  DenseTN la{A, M, K, M}, lb{B, N, K, N};
  launch_gemm_tn(la, lb, C, M, N, K, beta, s);
  return false;   // caller computes the bias gradient separately
}

// ---------------------------------------------------------------------------
// materialized im2col for C % 8 != 0 convs (honest RGB→4 first layer):
// the implicit packed loaders measured ~3× lower GEMM efficiency (mixed
// 8-B source parity halves DMA width), so these convs write the unroll
// ONCE as a [N·OH·OW, Kpad] bf16 image (Kpad = K rounded to 8, all
// chunks 16-B aligned) and run plain dense GEMMs — the image is shared
// by forward and wgrad within a step (pointer+geometry cache below).
// ---------------------------------------------------------------------------
// One thread per 8-B unit (4 bf16 = one C-quad of one (kh,kw) tap),
// unit-linear over the OUTPUT so every wave's 64 stores are 512
// contiguous bytes (the first cut mapped threads to rows — 8-B stores
// at a Kpad-stride, fully uncoalesced, and cost more than it saved).
__global__ void im2col_mat_k(const unsigned short* __restrict__ x,
                             unsigned short* __restrict__ out,
                             ConvDesc d, int Kpad, int nunits,
                             DivM mUrow, DivM mKWc, DivM mc4) {
  const unsigned u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u >= (unsigned)nunits) return;
  // u -> (row, unit j within the padded row)
  const unsigned r = divq(u, mUrow);
  const unsigned j = u - r * (unsigned)(Kpad >> 2);
  // j -> (kh, kw, c4); j >= K/4 is the alignment pad tail (zeros)
  const unsigned kh = divq(j, mKWc);
  const unsigned jr = j - kh * ((unsigned)d.KW * ((unsigned)d.C >> 2));
  const unsigned kw = divq(jr, mc4);
  const unsigned c4 = jr - kw * ((unsigned)d.C >> 2);
  // row -> (n, oh, ow)
  unsigned t2 = divq(r, d.mOW);
  unsigned ow = r - t2 * d.OW;
  unsigned n = divq(t2, d.mOH);
  unsigned oh = t2 - n * d.OH;
  const int ih = (int)oh * d.SY - d.PT + (int)kh;
  const int iw = (int)ow * d.SX - d.PL + (int)kw;
  const bool ok = (j < (unsigned)((d.KH * d.KW * d.C) >> 2)) &&
                  (unsigned)ih < (unsigned)d.H &&
                  (unsigned)iw < (unsigned)d.W;
  unsigned long long v =
      ok ? *(const unsigned long long*)(
               x + ((((long)n * d.H + ih) * d.W + iw) * d.C) + c4 * 4)
         : 0ull;
  *(unsigned long long*)(out + (long)u * 4) = v;
}

static unsigned short* im2col_ws_buf(size_t bytes) {
  static std::mutex mu;
  static unsigned short* ws = nullptr;
  static size_t cap = 0;
  std::lock_guard<std::mutex> lk(mu);
  if (bytes > cap) {
    if (ws) (void)hipFree(ws);
    cap = bytes;
    if (hipMalloc(&ws, cap) != hipSuccess) {
      ws = nullptr;
      cap = 0;
      throw std::runtime_error("im2col workspace alloc");
    }
  }
  return ws;
}

// Reuse cache: forward ALWAYS materializes (fresh minibatch each step);
// wgrad reuses the image iff the immediately-preceding materialize was
// for the same input+geometry (true within one step unless another
// C%8!=0 conv ran in between).  Under hipGraph capture the decision is
// baked once and stays correct: the captured sequence re-runs the
// forward's materialize node before the wgrad GEMM on every replay.
static const void* g_i2c_key_x = nullptr;
static long g_i2c_key_geom = 0;

static unsigned short* materialize_im2col(const unsigned short* x,
                                          const ConvDesc& d, int Kpad,
                                          hipStream_t s, bool allow_reuse) {
  const int rows = d.N * d.OH * d.OW;
  unsigned short* ws = im2col_ws_buf((size_t)rows * Kpad * 2);
  const long geom = ((long)d.N << 44) ^ ((long)d.H << 30) ^
                    ((long)d.C << 24) ^ ((long)d.KH << 18) ^
                    ((long)d.KW << 12) ^ ((long)d.SY << 8) ^ d.OW;
  if (allow_reuse && g_i2c_key_x == x && g_i2c_key_geom == geom)
    return ws;
  const long nunits_l = (long)rows * (Kpad >> 2);
  if (nunits_l >= (1l << 31))
    throw std::runtime_error("im2col image too large for one pass");
  const int nunits = (int)nunits_l;
  dim3 grid((nunits + 255) / 256), block(256);
  hipLaunchKernelGGL(im2col_mat_k, grid, block, 0, s, x, ws, d, Kpad,
                     nunits, make_divm((unsigned)(Kpad >> 2)),
                     make_divm((unsigned)(d.KW * (d.C >> 2))),
                     make_divm((unsigned)(d.C >> 2)));
  HIP_CHECK_LAST();
  g_i2c_key_x = x;
  g_i2c_key_geom = geom;
  return ws;
}

// ---------------------------------------------------------------------------
// strided dgrad via residue classes (one dense GEMM per (a, b) class +
// an interleave scatter) — see loaders.h DgradNTR.  The generic
// strided gather paid SY·SX× MFMA work (conv1-s4 dgrad measured ~21 ms
// standalone; AE-family stride-2 deconvs 4×).
// ---------------------------------------------------------------------------
__global__ void dgrad_scatter_k(const unsigned short* __restrict__ src,
                                unsigned short* __restrict__ ei,
                                ConvDesc d, DgradClass c, int nunits,
                                DivM mCu, DivM mWcu) {
  const unsigned u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u >= (unsigned)nunits) return;
  const unsigned row = divq(u, mCu);          // class-local pixel
  const unsigned cu = u - row * (unsigned)(d.C >> 2);
  unsigned t2 = divq(row, mWcu);
  unsigned iwr = row - t2 * c.Wc;
  unsigned n = divq(t2, c.mHc);
  unsigned ihr = t2 - n * c.Hc;
  const long dst = (((long)n * d.H + c.ih0 + ihr * d.SY) * d.W +
                    c.iw0 + iwr * d.SX) * d.C + cu * 4;
  *(unsigned long long*)(ei + dst) =
      src ? *(const unsigned long long*)(src + (long)u * 4) : 0ull;
}

static unsigned short* dgrad_ws_buf(size_t bytes) {
  static std::mutex mu;
  static unsigned short* ws = nullptr;
  static size_t cap = 0;
  std::lock_guard<std::mutex> lk(mu);
  if (bytes > cap) {
    if (ws) (void)hipFree(ws);
    cap = bytes;
    if (hipMalloc(&ws, cap) != hipSuccess) {
      ws = nullptr;
      cap = 0;
      throw std::runtime_error("dgrad residue workspace alloc");
    }
  }
  return ws;
}

static void conv_dgrad_residue(const unsigned short* eo,
                               const unsigned short* wdg,
                               unsigned short* ei,
                               const unsigned short* zero,
                               const ConvDesc& d, hipStream_t s) {
  const int N = d.C;
  unsigned short* wsb =
      dgrad_ws_buf((size_t)d.N * d.H * d.W * N * 2);
  long off = 0;
  struct ClassRec { DgradClass c; long off; long mc; bool has; };
  ClassRec recs[64];
  int nrec = 0;
  for (int a = 0; a < d.SY; ++a)
    for (int b = 0; b < d.SX; ++b) {
      DgradClass c;
      c.a = a;
      c.b = b;
      c.ih0 = ((a - d.PT) % d.SY + d.SY) % d.SY;
      c.iw0 = ((b - d.PL) % d.SX + d.SX) % d.SX;
      if (c.ih0 >= d.H || c.iw0 >= d.W) continue;
      c.Hc = (d.H - c.ih0 + d.SY - 1) / d.SY;
      c.Wc = (d.W - c.iw0 + d.SX - 1) / d.SX;
      c.KHc = a < d.KH ? (d.KH - a + d.SY - 1) / d.SY : 0;
      c.KWc = b < d.KW ? (d.KW - b + d.SX - 1) / d.SX : 0;
      c.ohb0 = (c.ih0 + d.PT - a) / d.SY;
      c.owb0 = (c.iw0 + d.PL - b) / d.SX;
      c.mWc = make_divm((unsigned)c.Wc);
      c.mHc = make_divm((unsigned)c.Hc);
      c.mKWc = make_divm((unsigned)(c.KWc > 0 ? c.KWc : 1));
      const long Mc = (long)d.N * c.Hc * c.Wc;
      const bool has = c.KHc > 0 && c.KWc > 0;
      if (has) {
        const int Kc = c.KHc * c.KWc * d.K;   // d.K % 8 == 0 upstream
        DgradNTR la{eo, zero, d, c};
        DenseDgradR lb{wdg, zero, d, c};
        launch_gemm_nt(la, lb, wsb + off, (const float*)nullptr,
                       (int)Mc, N, Kc, ACT_LINEAR, s);
      }
      recs[nrec++] = ClassRec{c, off, Mc, has};
      off += Mc * N;
    }
  for (int i = 0; i < nrec; ++i) {
    const ClassRec& r = recs[i];
    const int nunits = (int)(r.mc * (N >> 2));
    if (nunits <= 0) continue;
    dim3 grid((nunits + 255) / 256), block(256);
    hipLaunchKernelGGL(dgrad_scatter_k, grid, block, 0, s,
                       r.has ? wsb + r.off : nullptr, ei, d, r.c,
                       nunits, make_divm((unsigned)(N >> 2)),
                       r.c.mWc);
    HIP_CHECK_LAST();
  }
}

void conv_fwd_gemm8(const unsigned short*, const unsigned short*, void*,
                    bool, const float*, const unsigned short*,
                    const ConvDesc&, int, hipStream_t);
void conv_dgrad_gemm8(const unsigned short*, const unsigned short*,
                      unsigned short*, const unsigned short*,
                      const ConvDesc&, hipStream_t);

void conv_fwd_gemm(const unsigned short* x, const unsigned short* w,
                   void* out, bool out_f32, const float* bias,
                   const unsigned short* zero, const ConvDesc& d, int act,
                   hipStream_t s) {
  const int M = d.N * d.OH * d.OW;
  const int K = d.KH * d.KW * d.C;
  // The 8-phase 256²-tile core (gemm8.hip) is numerically verified but
  // measures at/below this 128² core on AlexNet/VGG shapes — its gain
  // needs the template's fine per-phase ds_read∥glds∥MFMA interleave
  // (guide: coarse phase-split −7..27%).  Kept opt-in for iteration.
  static const bool use8 = getenv("VELES_NT8") != nullptr;
  if (use8 && M >= 2048 && d.K >= 192 && (K & 7) == 0) {
    conv_fwd_gemm8(x, w, out, out_f32, bias, zero, d, act, s);
    return;
  }
  static const bool ntn8 = getenv("VELES_NO_NTN8") == nullptr;
  // N<=64 shapes take the deep core only at moderate M (VGG's 3.2M-row
  // conv2 measured -2.4% on it; CIFAR's sub-1M shapes +3%)
  if (ntn8 && (d.C & 7) == 0 &&
      (d.K > 64 ? d.K <= 96 : M < (1 << 20)) && d.K <= 96 &&
      (K & 7) == 0 && K >= 512 && M >= 4096) {
    Im2colNT la{x, zero, d};
    DenseNT lb{w, zero, K, K};
    if (out_f32)
      launch_gemm_ntn8(la, lb, (float*)out, bias, M, d.K, K, act, s);
    else
      launch_gemm_ntn8(la, lb, (unsigned short*)out, bias, M, d.K, K,
                       act, s);
    return;
  }
  if (d.C & 7) {
    if ((d.C & 3) == 0) {
      // honest C=4 path: one-shot materialized im2col + dense GEMM
      const int Kpad = (K + 7) & ~7;
      unsigned short* col = materialize_im2col(x, d, Kpad, s,
                                               /*allow_reuse=*/false);
      DenseNT la{col, zero, Kpad, Kpad};
      DenseNT lb{w, zero, K, K};        // k >= K reads the zero page
      if (out_f32)
        launch_gemm_nt(la, lb, (float*)out, bias, M, d.K, Kpad, act, s);
      else
        launch_gemm_nt(la, lb, (unsigned short*)out, bias, M, d.K, Kpad,
                       act, s);
      return;
    }
    Im2colNTP la{x, zero, d};
    DenseNT lb{w, zero, K, K};
    if (out_f32)
      launch_gemm_nt(la, lb, (float*)out, bias, M, d.K, K, act, s);
    else
      launch_gemm_nt(la, lb, (unsigned short*)out, bias, M, d.K, K, act,
                     s);
    return;
  }
  Im2colNT la{x, zero, d};
  DenseNT lb{w, zero, K, K};
  static const bool nt8d = getenv("VELES_NT8D") != nullptr;
  if (nt8d && d.K > 96 && (K & 7) == 0 && K >= 512 && M >= 8192) {
    if (out_f32)
      launch_gemm_nt8d(la, lb, (float*)out, bias, M, d.K, K, act, s);
    else
      launch_gemm_nt8d(la, lb, (unsigned short*)out, bias, M, d.K, K,
                       act, s);
    return;
  }
  if (out_f32)
    launch_gemm_nt(la, lb, (float*)out, bias, M, d.K, K, act, s);
  else
    launch_gemm_nt(la, lb, (unsigned short*)out, bias, M, d.K, K, act, s);
}

void conv_dgrad_gemm(const unsigned short* eo, const unsigned short* wdg,
                     unsigned short* ei, const unsigned short* zero,
                     const ConvDesc& d, hipStream_t s,
                     int act = ACT_LINEAR,
                     const unsigned short* dact = nullptr) {
  const int M = d.N * d.H * d.W;       // input pixels
  const int N = d.C;                   // input channels
  const int K = d.KH * d.KW * d.K;     // co-inner gather dim
  static const bool use8 = getenv("VELES_NT8") != nullptr;
  if (use8 && M >= 2048 && N >= 192 && (K & 7) == 0 && !dact) {
    conv_dgrad_gemm8(eo, wdg, ei, zero, d, s);
    return;
  }
  DenseNT lb{wdg, zero, K, K};         // wdg: [C][KH][KW][K]
  if (d.SY == 1 && d.SX == 1) {
    DgradNT1 la{eo, zero, d};
    static const bool ntn8 = getenv("VELES_NO_NTN8") == nullptr;
    if (ntn8 && (N > 64 ? N <= 96 : M < (1 << 20)) && N <= 96 &&
        (N & 7) == 0 && (K & 7) == 0 && K >= 512 && M >= 4096) {
      launch_gemm_ntn8(la, lb, ei, (const float*)nullptr, M, N, K, act,
                       s, dact);
      return;
    }
    static const bool nt8d = getenv("VELES_NT8D") != nullptr;
    if (nt8d && N > 96 && (K & 7) == 0 && K >= 512 && M >= 8192) {
      launch_gemm_nt8d(la, lb, ei, (const float*)nullptr, M, N, K, act,
                       s, dact);
      return;
    }
    launch_gemm_nt(la, lb, ei, (const float*)nullptr, M, N, K, act, s,
                   dact);
    return;
  }
  if ((d.K & 7) == 0 && (d.C & 3) == 0 && d.SY < 8 && d.SX < 8 &&
      !dact) {
    conv_dgrad_residue(eo, wdg, ei, zero, d, s);
    return;
  }
  DgradNT la{eo, zero, d};
  launch_gemm_nt(la, lb, ei, (const float*)nullptr, M, N, K, act, s,
                 dact);
}

bool conv_wgrad_gemm(const unsigned short* eo, const unsigned short* x,
                     float* gw, const ConvDesc& d, float beta,
                     hipStream_t s, float* bias_out) {
  const int Kdim = d.N * d.OH * d.OW;
  const int M = d.K;                   // n_kernels
  const int N = d.KH * d.KW * d.C;
  if ((M & 7) == 0 && (d.C & 7) == 0) {
    DenseTNv la{eo, d.K, Kdim, d.K};
    Im2colTNv lb{x, d, Kdim, N};
    return launch_gemm_tn2(la, lb, gw, M, N, Kdim, beta, s, bias_out);
  }
  if ((M & 7) == 0 && (d.C & 3) == 0) {
    // reuse (or rebuild) the forward's materialized im2col image and
    // run a plain dense K-major GEMM — the image IS K-major for TN
    const int Kpad = (N + 7) & ~7;
    unsigned short* col = materialize_im2col(x, d, Kpad, s,
                                             /*allow_reuse=*/true);
    DenseTNv la{eo, d.K, Kdim, d.K};
    DenseTNv lb{col, Kpad, Kdim, Kpad};  // pad cols are zero-filled
    return launch_gemm_tn2(la, lb, gw, M, N, Kdim, beta, s, bias_out);
  }
  DenseTN la{eo, d.K, Kdim, d.K};
  Im2colTN lb{x, d, Kdim, N};
  launch_gemm_tn(la, lb, gw, M, N, Kdim, beta, s);
  return false;   // caller computes the bias gradient separately
}
