// Per-lane global source-address generators for the NT GEMM cores
// (shared by the 128x128 2-barrier core in gemm.hip and the 256x256
// 8-phase core in gemm8.hip).
#pragma once
#include "common.h"

// Two-level address API (all fast loaders): the ROW decomposition of a
// staged chunk is loop-invariant across K-tiles, so the cores compute
// `RowC rowc(row)` once per thread before the K-loop and per tile only
// the K side runs: `KC kc(k8)` (shared by every chunk with that
// k-octet) + `addr_rk(rowc, kc)` (adds + bounds only).  This moved the
// magic-division work out of the staging hot path entirely.

// dense row-major [rows][ld], K-innermost; k past Klim reads the zero page
struct DenseNT {
  static constexpr bool MAY_SLOW = false;   // addr() never nullptr
  static constexpr bool N96_OK = true;
  const unsigned short* base;
  const unsigned short* zero;
  long ld;
  int Klim;
  __device__ const void* addr(int row, int k8) const {
    int k = k8 * 8;
    return k < Klim ? (const void*)(base + (long)row * ld + k)
                    : (const void*)zero;
  }
  // single-element access for the K%8 tail chunk
  __device__ unsigned short elem(int row, int k) const {
    return k < Klim ? base[(long)row * ld + k] : (unsigned short)0;
  }
  struct RowC { const unsigned short* rowbase; };
  struct KC { int k; };
  __device__ RowC rowc(int row) const {
    return RowC{base + (long)row * ld};
  }
  __device__ KC kc(int k8) const { return KC{k8 * 8}; }
  __device__ const void* addr_rk(const RowC& rc, const KC& c) const {
    return c.k < Klim ? (const void*)(rc.rowbase + c.k)
                      : (const void*)zero;
  }
};

// implicit im2col over NHWC input: row=(n,oh,ow), kdim=(kh,kw,c) c-inner.
// Requires C % 8 == 0 so each 16-B chunk stays inside one pixel's C-run
// (Im2colNTP below handles C%4).
struct Im2colNT {
  static constexpr bool MAY_SLOW = false;
  static constexpr bool N96_OK = true;
  const unsigned short* x;
  const unsigned short* zero;
  ConvDesc d;
  __device__ const void* addr(int row, int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mC);
    unsigned c = k - t * d.C;
    unsigned kh = divq(t, d.mKW);
    unsigned kw = t - kh * d.KW;
    unsigned t2 = divq((unsigned)row, d.mOW);
    unsigned ow = (unsigned)row - t2 * d.OW;
    unsigned n = divq(t2, d.mOH);
    unsigned oh = t2 - n * d.OH;
    int ih = (int)oh * d.SY - d.PT + (int)kh;
    int iw = (int)ow * d.SX - d.PL + (int)kw;
    if ((int)kh >= d.KH || (unsigned)ih >= (unsigned)d.H ||
        (unsigned)iw >= (unsigned)d.W)
      return zero;
    return x + (((long)n * d.H + ih) * d.W + iw) * d.C + c;
  }
  struct RowC { long off0; int ih0, iw0; };
  struct KC { int c, kw, kh; };
  __device__ RowC rowc(int row) const {
    unsigned t2 = divq((unsigned)row, d.mOW);
    unsigned ow = (unsigned)row - t2 * d.OW;
    unsigned n = divq(t2, d.mOH);
    unsigned oh = t2 - n * d.OH;
    int ih0 = (int)oh * d.SY - d.PT;
    int iw0 = (int)ow * d.SX - d.PL;
    return RowC{(((long)n * d.H + ih0) * d.W + iw0) * d.C, ih0, iw0};
  }
  __device__ KC kc(int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mC);
    unsigned c = k - t * d.C;
    unsigned kh = divq(t, d.mKW);
    unsigned kw = t - kh * d.KW;
    return KC{(int)c, (int)kw, (int)kh};
  }
  __device__ const void* addr_rk(const RowC& rc, const KC& c) const {
    if (c.kh >= d.KH || (unsigned)(rc.ih0 + c.kh) >= (unsigned)d.H ||
        (unsigned)(rc.iw0 + c.kw) >= (unsigned)d.W)
      return zero;
    return x + rc.off0 + ((long)c.kh * d.W + c.kw) * d.C + c.c;
  }
  __device__ unsigned short elem(int row, int k) const {
    // only hit when K%8 != 0 (conv K-dims are C-padded so never here)
    int c = k % d.C, t = k / d.C;
    int kw = t % d.KW, kh = t / d.KW;
    int ow = row % d.OW;
    t = row / d.OW;
    int oh = t % d.OH, n = t / d.OH;
    int ih = oh * d.SY - d.PT + kh;
    int iw = ow * d.SX - d.PL + kw;
    if (kh >= d.KH || (unsigned)ih >= (unsigned)d.H ||
        (unsigned)iw >= (unsigned)d.W)
      return 0;
    return x[(((long)n * d.H + ih) * d.W + iw) * d.C + c];
  }
};

// Packed im2col for C % 4 == 0 (RGB padded to 4, not 8): a 16-B chunk
// spans several contiguous (kw,c) pairs of one kh row.  addr() returns
// nullptr for chunks where the straight DMA is illegal — kh-row
// crossing, horizontal padding in play, or a non-8-B-aligned source
// (8-B-aligned glds-16 is legal on gfx950: tools/probe_glds_align.hip)
// — and the staging loop per-element-loads just those.  A separate
// TYPE so the fallback machinery costs the C%8 kernels nothing.
struct Im2colNTP {
  static constexpr bool MAY_SLOW = true;
  static constexpr bool N96_OK = true;
  const unsigned short* x;
  const unsigned short* zero;
  ConvDesc d;
  __device__ const void* addr(int row, int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mC);
    unsigned c = k - t * d.C;
    unsigned kh = divq(t, d.mKW);
    unsigned kw = t - kh * d.KW;
    unsigned t2 = divq((unsigned)row, d.mOW);
    unsigned ow = (unsigned)row - t2 * d.OW;
    unsigned n = divq(t2, d.mOH);
    unsigned oh = t2 - n * d.OH;
    int ih = (int)oh * d.SY - d.PT + (int)kh;
    int iw = (int)ow * d.SX - d.PL + (int)kw;
    const int rowlen = d.KW * d.C;
    const int kinrow = (int)(kw * d.C + c);    // k % (KW*C)
    if (kinrow + 8 > rowlen || (d.PT | d.PL))
      return nullptr;                  // kh-cross or padded: slow path
    if ((int)kh >= d.KH || (unsigned)ih >= (unsigned)d.H)
      return zero;
    long off = (((long)n * d.H + ih) * d.W + iw) * d.C + c;
    return (off & 3) ? nullptr : (const void*)(x + off);
  }
  __device__ unsigned short elem(int row, int k) const {
    int c = k % d.C, t = k / d.C;
    int kw = t % d.KW, kh = t / d.KW;
    int ow = row % d.OW;
    t = row / d.OW;
    int oh = t % d.OH, n = t / d.OH;
    int ih = oh * d.SY - d.PT + kh;
    int iw = ow * d.SX - d.PL + kw;
    if (kh >= d.KH || (unsigned)ih >= (unsigned)d.H ||
        (unsigned)iw >= (unsigned)d.W)
      return 0;
    return x[(((long)n * d.H + ih) * d.W + iw) * d.C + c];
  }

  // passthrough two-level API (cores only take it on !MAY_SLOW paths,
  // but the templates must compile for every loader)
  struct RowC { int row; };
  struct KC { int k8; };
  __device__ RowC rowc(int row) const { return RowC{row}; }
  __device__ KC kc(int k8) const { return KC{k8}; }
  __device__ const void* addr_rk(const RowC& rc, const KC& c) const {
    return addr(rc.row, c.k8);
  }

  // refused-chunk path: a kh-crossing octet is exactly two contiguous
  // 8-B-aligned runs (C=4: 4+4 elements) — two vector loads instead of
  // eight guarded scalars (per-element vmcnt serialization measured
  // ~10-30× a fast chunk).  Padded convs take the scalar path.
  __device__ void slowload(int row, int kc0, unsigned short* v) const {
    const int rowlen = d.KW * d.C;
    const int pos = kc0 % rowlen;
    if ((d.PT | d.PL) || rowlen - pos != 4 || (d.C & 3)) {
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = elem(row, kc0 + j);
      return;
    }
    const int kh = kc0 / rowlen;         // run 1: last 4 elems of row kh
    const int ow = row % d.OW;
    const int t = row / d.OW;
    const int oh = t % d.OH, n = t / d.OH;
    const int ih = oh * d.SY + kh;       // PT == 0 here
    const int iw = ow * d.SX + (pos / d.C);
    unsigned long long r1 = 0, r2 = 0;
    if (kh < d.KH && ih < d.H)
      r1 = *(const unsigned long long*)(
          x + (((long)n * d.H + ih) * d.W + iw) * d.C + (pos % d.C));
    if (kh + 1 < d.KH && ih + 1 < d.H)
      r2 = *(const unsigned long long*)(
          x + (((long)n * d.H + ih + 1) * d.W + ow * d.SX) * d.C);
    *(unsigned long long*)v = r1;
    *((unsigned long long*)v + 1) = r2;
  }
};

// conv dgrad gather over NHWC err_output: row=(n,ih,iw) of the FORWARD
// input, kdim=(kh,kw,co) co-inner; strided positions read the zero page.
struct DgradNT {
  static constexpr bool MAY_SLOW = false;
  // measured: the 96-wide narrow tile LOSES ~6% on the dgrad gather
  static constexpr bool N96_OK = false;
  const unsigned short* eo;
  const unsigned short* zero;
  ConvDesc d;
  __device__ const void* addr(int row, int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mK);
    unsigned co = k - t * d.K;
    unsigned kh = divq(t, d.mKW);
    unsigned kw = t - kh * d.KW;
    unsigned t2 = divq((unsigned)row, d.mW);
    unsigned iw = (unsigned)row - t2 * d.W;
    unsigned n = divq(t2, d.mH);
    unsigned ih = t2 - n * d.H;
    int ohn = (int)ih + d.PT - (int)kh;
    int own = (int)iw + d.PL - (int)kw;
    unsigned oh = divq((unsigned)ohn, d.mSY);
    unsigned ow = divq((unsigned)own, d.mSX);
    if ((int)kh >= d.KH || ohn < 0 || own < 0 ||
        ohn != (int)(oh * d.SY) || own != (int)(ow * d.SX))
      return zero;
    if ((int)oh >= d.OH || (int)ow >= d.OW) return zero;
    return eo + (((long)n * d.OH + oh) * d.OW + ow) * d.K + co;
  }
  struct RowC { long nbase; int oh0, ow0; };
  struct KC { int co, kw, kh; };
  __device__ RowC rowc(int row) const {
    unsigned t2 = divq((unsigned)row, d.mW);
    unsigned iw = (unsigned)row - t2 * d.W;
    unsigned n = divq(t2, d.mH);
    unsigned ih = t2 - n * d.H;
    return RowC{(long)n * d.OH * d.OW * d.K, (int)ih + d.PT,
                (int)iw + d.PL};
  }
  __device__ KC kc(int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mK);
    unsigned co = k - t * d.K;
    unsigned kh = divq(t, d.mKW);
    unsigned kw = t - kh * d.KW;
    return KC{(int)co, (int)kw, (int)kh};
  }
  __device__ const void* addr_rk(const RowC& rc, const KC& c) const {
    int ohn = rc.oh0 - c.kh, own = rc.ow0 - c.kw;
    unsigned oh = divq((unsigned)ohn, d.mSY);
    unsigned ow = divq((unsigned)own, d.mSX);
    if (c.kh >= d.KH || ohn < 0 || own < 0 ||
        ohn != (int)(oh * d.SY) || own != (int)(ow * d.SX) ||
        (int)oh >= d.OH || (int)ow >= d.OW)
      return zero;
    return eo + rc.nbase + ((long)oh * d.OW + ow) * d.K + c.co;
  }
  __device__ unsigned short elem(int row, int k) const {
    int co = k % d.K, t = k / d.K;
    int kw = t % d.KW, kh = t / d.KW;
    int iw = row % d.W;
    t = row / d.W;
    int ih = t % d.H, n = t / d.H;
    int ohn = ih + d.PT - kh;
    int own = iw + d.PL - kw;
    if (kh >= d.KH || ohn < 0 || own < 0 || ohn % d.SY || own % d.SX)
      return 0;
    int oh = ohn / d.SY, ow = own / d.SX;
    if (oh >= d.OH || ow >= d.OW) return 0;
    return eo[(((long)n * d.OH + oh) * d.OW + ow) * d.K + co];
  }
};

// stride-1 dgrad specialization: SY == SX == 1 removes the per-address
// stride divisibility checks and divisions of the generic DgradNT.
struct DgradNT1 {
  static constexpr bool MAY_SLOW = false;
  static constexpr bool N96_OK = false;
  const unsigned short* eo;
  const unsigned short* zero;
  ConvDesc d;
  __device__ const void* addr(int row, int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mK);
    unsigned co = k - t * d.K;
    unsigned kh = divq(t, d.mKW);
    unsigned kw = t - kh * d.KW;
    unsigned t2 = divq((unsigned)row, d.mW);
    unsigned iw = (unsigned)row - t2 * d.W;
    unsigned n = divq(t2, d.mH);
    unsigned ih = t2 - n * d.H;
    int oh = (int)ih + d.PT - (int)kh;
    int ow = (int)iw + d.PL - (int)kw;
    if ((int)kh >= d.KH || (unsigned)oh >= (unsigned)d.OH ||
        (unsigned)ow >= (unsigned)d.OW)
      return zero;
    return eo + (((long)n * d.OH + oh) * d.OW + ow) * d.K + co;
  }
  struct RowC { long off0; int oh0, ow0; };
  struct KC { int co, kw, kh; };
  __device__ RowC rowc(int row) const {
    unsigned t2 = divq((unsigned)row, d.mW);
    unsigned iw = (unsigned)row - t2 * d.W;
    unsigned n = divq(t2, d.mH);
    unsigned ih = t2 - n * d.H;
    int oh0 = (int)ih + d.PT, ow0 = (int)iw + d.PL;
    return RowC{(((long)n * d.OH + oh0) * d.OW + ow0) * d.K, oh0, ow0};
  }
  __device__ KC kc(int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mK);
    unsigned co = k - t * d.K;
    unsigned kh = divq(t, d.mKW);
    unsigned kw = t - kh * d.KW;
    return KC{(int)co, (int)kw, (int)kh};
  }
  __device__ const void* addr_rk(const RowC& rc, const KC& c) const {
    if (c.kh >= d.KH || (unsigned)(rc.oh0 - c.kh) >= (unsigned)d.OH ||
        (unsigned)(rc.ow0 - c.kw) >= (unsigned)d.OW)
      return zero;
    return eo + rc.off0 - ((long)c.kh * d.OW + c.kw) * d.K + c.co;
  }
  __device__ unsigned short elem(int row, int k) const {
    int co = k % d.K, t = k / d.K;
    int kw = t % d.KW, kh = t / d.KW;
    int iw = row % d.W;
    t = row / d.W;
    int ih = t % d.H, n = t / d.H;
    int oh = ih + d.PT - kh;
    int ow = iw + d.PL - kw;
    if (kh >= d.KH || (unsigned)oh >= (unsigned)d.OH ||
        (unsigned)ow >= (unsigned)d.OW)
      return 0;
    return eo[(((long)n * d.OH + oh) * d.OW + ow) * d.K + co];
  }
};

static __device__ __forceinline__ int xcd_tile(int b, int ntiles) {
  // blocks go to XCD (b % 8); remap so tiles of one XCD are contiguous
  // (per-XCD L2 reuse of the shared B/A panels) — speed-only, any
  // mapping is correct.
  if (ntiles < 64) return b;
  const int per = ntiles >> 3;            // tiles per XCD (floor)
  const int cut = per << 3;
  if (b >= cut) return b;                 // remainder keeps identity
  return (b & 7) * per + (b >> 3);
}


// ---------------------------------------------------------------------------
// Residue-class strided dgrad (SY or SX > 1): the generic DgradNT pays
// SY·SX× MFMA work on taps whose stride check fails (zero-page reads).
// Decomposing input pixels by (ih+PT) % SY, (iw+PL) % SX makes every
// tap of a class DENSE: kh = a + kh'·SY, kw = b + kw'·SX, and the
// oh/ow arithmetic collapses to adds.  One GEMM per class into a
// contiguous buffer; an interleave kernel scatters classes into ei.
// ---------------------------------------------------------------------------
struct DgradClass {
  int ih0, iw0;        // first input row/col of the class
  int Hc, Wc;          // class spatial extent
  int KHc, KWc;        // taps per class (kh = a + kh'*SY, ...)
  int a, b;            // residues
  int ohb0, owb0;      // (ih0+PT-a)/SY, (iw0+PL-b)/SX
  DivM mWc, mHc, mKWc; // class-local magic divisors
};

struct DgradNTR {
  static constexpr bool MAY_SLOW = false;
  static constexpr bool N96_OK = false;
  const unsigned short* eo;
  const unsigned short* zero;
  ConvDesc d;
  DgradClass c;
  struct RowC { long off0; int ohm, owm; };
  struct KC { int co, kwp, khp; };
  __device__ RowC rowc(int row) const {
    unsigned t2 = divq((unsigned)row, c.mWc);
    unsigned iwr = (unsigned)row - t2 * c.Wc;
    unsigned n = divq(t2, c.mHc);
    unsigned ihr = t2 - n * c.Hc;
    const int ohm = c.ohb0 + (int)ihr;
    const int owm = c.owb0 + (int)iwr;
    return RowC{(((long)n * d.OH + ohm) * d.OW + owm) * d.K, ohm, owm};
  }
  __device__ KC kc(int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mK);
    unsigned co = k - t * d.K;
    unsigned khp = divq(t, c.mKWc);
    unsigned kwp = t - khp * c.KWc;
    return KC{(int)co, (int)kwp, (int)khp};
  }
  __device__ const void* addr_rk(const RowC& rc, const KC& kk) const {
    const int oh = rc.ohm - kk.khp;
    const int ow = rc.owm - kk.kwp;
    // khp >= KHc <=> k >= Kc: the GEMM K is padded to a BK multiple
    // and the tail must read zeros (decoded khp would alias to VALID
    // oh values at high rows otherwise)
    if (kk.khp >= c.KHc || (unsigned)oh >= (unsigned)d.OH ||
        (unsigned)ow >= (unsigned)d.OW)
      return zero;
    return eo + rc.off0 - ((long)kk.khp * d.OW + kk.kwp) * d.K + kk.co;
  }
  __device__ const void* addr(int row, int k8) const {
    return addr_rk(rowc(row), kc(k8));
  }
  __device__ unsigned short elem(int row, int k) const {  // K%8 never
    return 0;
  }
};

// W-dgrad image [C][KH][KW][K] restricted to a class's taps: a 16-B
// chunk is 8 consecutive co at fixed (kh, kw) — contiguous (K % 8 == 0)
struct DenseDgradR {
  static constexpr bool MAY_SLOW = false;
  static constexpr bool N96_OK = false;
  const unsigned short* w;
  const unsigned short* zero;
  ConvDesc d;
  DgradClass c;
  struct RowC { long rbase; };
  struct KC { int off; };
  __device__ RowC rowc(int row) const {
    return RowC{(long)row * d.KH * d.KW * d.K};
  }
  __device__ KC kc(int k8) const {
    unsigned k = (unsigned)k8 * 8;
    unsigned t = divq(k, d.mK);
    unsigned co = k - t * d.K;
    unsigned khp = divq(t, c.mKWc);
    unsigned kwp = t - khp * c.KWc;
    if ((int)khp >= c.KHc) return KC{-1};       // padded K tail
    const int kh = c.a + (int)khp * d.SY;
    const int kw = c.b + (int)kwp * d.SX;
    return KC{(int)(((long)kh * d.KW + kw) * d.K + co)};
  }
  __device__ const void* addr_rk(const RowC& rc, const KC& kk) const {
    return kk.off < 0 ? (const void*)zero : (const void*)(w + rc.rbase + kk.off);
  }
  __device__ const void* addr(int row, int k8) const {
    return addr_rk(rowc(row), kc(k8));
  }
  __device__ unsigned short elem(int row, int k) const { return 0; }
};
