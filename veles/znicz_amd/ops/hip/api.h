// Host-safe shared types for the _mzk kernel set (no device code) —
// included by both bindings.cpp (g++) and the .hip kernels (hipcc).
#pragma once

enum Act { ACT_LINEAR = 0, ACT_TANH = 1, ACT_RELU = 2,
           ACT_STRICT_RELU = 3, ACT_SIGMOID = 4 };

// Magic-number unsigned division: q = (u32)(((u64)n * mul) >> shift),
// exact for every 0 <= n < 2^31 and divisor d >= 1.  A runtime-divisor
// `/` compiles to ~30 VALU instructions on gfx950; this is mul_lo +
// mul_hi + 64-b shift (~4) — the GEMM loaders issue several per staged
// 16-B chunk, directly competing with MFMA on the same SIMDs.
struct DivM {
  unsigned mul, shift;
};

inline DivM make_divm(unsigned d) {
  if ((d & (d - 1)) == 0) {            // power of two (incl. 1)
    unsigned z = 0;
    while ((1u << z) < d) ++z;
    return DivM{0x80000000u, 31 + z};
  }
  unsigned lg = 31 - __builtin_clz(d);
  unsigned shift = 32 + lg;
  unsigned long long m = ((1ull << shift) + d - 1) / d;   // ceil
  return DivM{(unsigned)m, shift};
}

struct ConvDesc {
  int N, H, W, C;      // input (NHWC)
  int K, KH, KW;       // kernels
  int OH, OW;          // output spatial
  int SY, SX;          // stride
  int PT, PL;          // top/left padding
  // precomputed magic divisors for the im2col/dgrad loaders
  DivM mC, mKW, mOW, mOH, mK, mSY, mSX, mW, mH;
};

inline void finish_conv_desc(ConvDesc& d) {
  d.mC = make_divm(d.C);
  d.mKW = make_divm(d.KW);
  d.mOW = make_divm(d.OW);
  d.mOH = make_divm(d.OH);
  d.mK = make_divm(d.K);
  d.mSY = make_divm(d.SY);
  d.mSX = make_divm(d.SX);
  d.mW = make_divm(d.W);
  d.mH = make_divm(d.H);
}

enum PoolMode { POOL_MAX = 0, POOL_AVG = 1, POOL_MAXABS = 2,
                POOL_STOCHASTIC = 3 };

struct PoolDesc {
  int N, H, W, C;
  int KY, KX, SY, SX;
  int OH, OW;
  unsigned long long seed;   // for stochastic pooling
};
