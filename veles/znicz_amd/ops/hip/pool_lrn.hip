// Pooling + cross-channel LRN kernels (NHWC bf16).
// Replaces reference znicz/ocl/{pooling.cl, gd_pooling.cl,
// normalization.cl} — redesigned: gather-based backward (no atomics),
// c-innermost layout so consecutive lanes stream consecutive channels.
#include "common.h"

// -- forward: lanes cover channels via pow2 mask (no div/mod per elem),
// rows (n, oh, ow) decoded once per row -------------------------------------
template <int MODE>
__device__ __forceinline__ void pool_fwd_one(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    unsigned char* __restrict__ offs, const PoolDesc& d, long idx, int n,
    int ih0, int iw0, int c);

template <int MODE>
// offsets are u8 RELATIVE window codes (ky*KX + kx; 255 = none) —
// int32 absolute indices cost 4× the offset traffic on both passes,
// and the backward can reconstruct the absolute position from its own
// (ih, iw) arithmetic.  AVG mode stores the window tap COUNT instead.
__global__ void pool_fwd_k(const unsigned short* __restrict__ x,
                           unsigned short* __restrict__ y,
                           unsigned char* __restrict__ offs, PoolDesc d,
                           int lg) {
  const int C2 = 1 << lg;
  const int c0 = threadIdx.x & (C2 - 1);
  const int rin = threadIdx.x >> lg;
  const int rpb = blockDim.x >> lg;
  const unsigned rows = (unsigned)d.N * d.OH * d.OW;
  const bool vec8 = (d.C & 7) == 0;
  for (unsigned row = blockIdx.x * rpb + rin; row < rows;
       row += gridDim.x * rpb) {
    unsigned t = row / d.OW;
    const int ow = row - t * d.OW;
    const int n = t / d.OH;
    const int oh = t - n * d.OH;
    const int ih0 = oh * d.SY, iw0 = ow * d.SX;
    if (vec8) {
      // octet-per-thread: one b128 window load serves 8 channels
      const int noct = d.C >> 3;
      for (int oc = c0; oc < noct; oc += C2) {
        const int cb = oc * 8;
        const long idx0 = (long)row * d.C + cb;
        float best[8], sum[8];
        int bestoff[8];
        int count = 0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          best[j] = 0.f;
          sum[j] = 0.f;
          bestoff[j] = -1;
        }
        float besta[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) besta[j] = -1.f;
        float st_t[8], st_a[8];
        if (MODE == POOL_STOCHASTIC) {
          float possum[8] = {};
          for (int ky = 0; ky < d.KY; ++ky) {
            int ih = ih0 + ky;
            if (ih >= d.H) break;
            for (int kx = 0; kx < d.KX; ++kx) {
              int iw = iw0 + kx;
              if (iw >= d.W) break;
              unsigned short v[8];
              *(short8*)v = *(const short8*)(
                  x + (((long)n * d.H + ih) * d.W + iw) * d.C + cb);
#pragma unroll
              for (int j = 0; j < 8; ++j)
                possum[j] += fmaxf(bf2f(v[j]), 0.f);
            }
          }
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            st_t[j] = possum[j] * uniform01(
                d.seed, (unsigned long long)(idx0 + j));
            st_a[j] = 0.f;
          }
        }
        for (int ky = 0; ky < d.KY; ++ky) {
          int ih = ih0 + ky;
          if (ih >= d.H) break;
          for (int kx = 0; kx < d.KX; ++kx) {
            int iw = iw0 + kx;
            if (iw >= d.W) break;
            long off = (((long)n * d.H + ih) * d.W + iw) * d.C + cb;
            const int rcode = ky * d.KX + kx;
            unsigned short v[8];
            *(short8*)v = *(const short8*)(x + off);
            ++count;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              float vj = bf2f(v[j]);
              if (MODE == POOL_AVG) {
                sum[j] += vj;
              } else if (MODE == POOL_MAX) {
                if (bestoff[j] < 0 || vj > best[j]) {
                  best[j] = vj;
                  bestoff[j] = rcode;
                }
              } else if (MODE == POOL_MAXABS) {
                float aj = fabsf(vj);
                if (bestoff[j] < 0 || aj > besta[j]) {
                  best[j] = vj;
                  besta[j] = aj;
                  bestoff[j] = rcode;
                }
              } else {
                float pv = fmaxf(vj, 0.f);
                st_a[j] += pv;
                if (bestoff[j] < 0 && st_a[j] > st_t[j] && pv > 0.f) {
                  best[j] = vj;
                  bestoff[j] = rcode;
                }
              }
            }
          }
        }
        unsigned short o8[8];
        unsigned char i8[8];
        if (MODE == POOL_AVG) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            o8[j] = f2bf(sum[j] / max(count, 1));
            i8[j] = (unsigned char)count;
          }
        } else {
          if (MODE == POOL_STOCHASTIC) {
            // all-nonpositive windows fall back to max
            for (int ky = 0; ky < d.KY; ++ky) {
              int ih = ih0 + ky;
              if (ih >= d.H) break;
              for (int kx = 0; kx < d.KX; ++kx) {
                int iw = iw0 + kx;
                if (iw >= d.W) break;
                long off = (((long)n * d.H + ih) * d.W + iw) * d.C + cb;
                const int rcode = ky * d.KX + kx;
                unsigned short v[8];
                *(short8*)v = *(const short8*)(x + off);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                  float vj = bf2f(v[j]);
                  if (st_a[j] <= 0.f &&
                      (bestoff[j] < 0 || vj > best[j])) {
                    best[j] = vj;
                    bestoff[j] = rcode;
                  }
                }
              }
            }
          }
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            o8[j] = f2bf(best[j]);
            i8[j] = bestoff[j] < 0 ? 255 : (unsigned char)bestoff[j];
          }
        }
        *(short8*)(y + idx0) = *(short8*)o8;
        if (offs)
          *(unsigned long long*)(offs + idx0) =
              *(unsigned long long*)i8;
      }
    } else {
      for (int c = c0; c < d.C; c += C2) {
        const long idx = (long)row * d.C + c;
        pool_fwd_one<MODE>(x, y, offs, d, idx, n, ih0, iw0, c);
      }
    }
  }
}

template <int MODE>
__device__ __forceinline__ void pool_fwd_one(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    unsigned char* __restrict__ offs, const PoolDesc& d, long idx, int n,
    int ih0, int iw0, int c) {
  float best = 0.f, besta = -1.f, sum = 0.f;
  int bestoff = -1, count = 0;
  float stoch_target = 0.f, stoch_acc = 0.f;
  if (MODE == POOL_STOCHASTIC) {
    // two-pass: first the sum of positive values, then pick
    float possum = 0.f;
    for (int ky = 0; ky < d.KY; ++ky) {
      int ih = ih0 + ky;
      if (ih >= d.H) break;
      for (int kx = 0; kx < d.KX; ++kx) {
        int iw = iw0 + kx;
        if (iw >= d.W) break;
        possum += fmaxf(bf2f(x[(((long)n * d.H + ih) * d.W + iw) * d.C + c]),
                        0.f);
      }
    }
    stoch_target = possum * uniform01(d.seed, (unsigned long long)idx);
  }
  for (int ky = 0; ky < d.KY; ++ky) {
    int ih = ih0 + ky;
    if (ih >= d.H) break;
    for (int kx = 0; kx < d.KX; ++kx) {
      int iw = iw0 + kx;
      if (iw >= d.W) break;
      long off = (((long)n * d.H + ih) * d.W + iw) * d.C + c;
      const int rcode = ky * d.KX + kx;
      float v = bf2f(x[off]);
      ++count;
      if (MODE == POOL_AVG) {
        sum += v;
      } else if (MODE == POOL_MAX) {
        if (bestoff < 0 || v > best) { best = v; bestoff = rcode; }
      } else if (MODE == POOL_MAXABS) {
        float a = fabsf(v);
        if (bestoff < 0 || a > besta) { best = v; besta = a;
                                        bestoff = rcode; }
      } else {  // stochastic: first cell where running pos-sum crosses
        float pv = fmaxf(v, 0.f);
        stoch_acc += pv;
        if (bestoff < 0 && (stoch_acc > stoch_target && pv > 0.f)) {
          best = v;
          bestoff = rcode;
        }
      }
    }
  }
  if (MODE == POOL_AVG) {
    y[idx] = f2bf(sum / max(count, 1));
    if (offs) offs[idx] = (unsigned char)count;  // divisor for backward
  } else {
    if (MODE == POOL_STOCHASTIC && bestoff < 0) {
      // all-nonpositive window: fall back to max
      for (int ky = 0; ky < d.KY; ++ky) {
        int ih = ih0 + ky;
        if (ih >= d.H) break;
        for (int kx = 0; kx < d.KX; ++kx) {
          int iw = iw0 + kx;
          if (iw >= d.W) break;
          float v = bf2f(x[(((long)n * d.H + ih) * d.W + iw) * d.C + c]);
          if (bestoff < 0 || v > best) { best = v;
                                         bestoff = ky * d.KX + kx; }
        }
      }
    }
    y[idx] = f2bf(best);
    if (offs) offs[idx] = bestoff < 0 ? 255 : (unsigned char)bestoff;
  }
}

// -- backward: gather over input elements (no atomics); same pow2
// lane->channel mapping, row (n, ih, iw) decoded once per row ---------------
template <int MODE>
__global__ void pool_bwd_k(const unsigned short* __restrict__ ey,
                           unsigned short* __restrict__ ex,
                           const unsigned char* __restrict__ offs,
                           PoolDesc d, int lg) {
  const int C2 = 1 << lg;
  const int c0 = threadIdx.x & (C2 - 1);
  const int rin = threadIdx.x >> lg;
  const int rpb = blockDim.x >> lg;
  const unsigned rows = (unsigned)d.N * d.H * d.W;
  const bool vec8 = (d.C & 7) == 0;
  for (unsigned row = blockIdx.x * rpb + rin; row < rows;
       row += gridDim.x * rpb) {
    unsigned t = row / d.W;
    const int iw = row - t * d.W;
    const int n = t / d.H;
    const int ih = t - n * d.H;
    const int oy0 = (ih - d.KY + 1 <= 0) ? 0 : (ih - d.KY) / d.SY + 1;
    const int oy1 = min(d.OH - 1, ih / d.SY);
    const int ox0 = (iw - d.KX + 1 <= 0) ? 0 : (iw - d.KX) / d.SX + 1;
    const int ox1 = min(d.OW - 1, iw / d.SX);
    if (vec8) {
      const int noct = d.C >> 3;
      for (int oc = c0; oc < noct; oc += C2) {
        const int cb = oc * 8;
        const long idx0 = (long)row * d.C + cb;
        float g[8] = {};
        for (int oy = oy0; oy <= oy1; ++oy)
          for (int ox = ox0; ox <= ox1; ++ox) {
            long o = (((long)n * d.OH + oy) * d.OW + ox) * d.C + cb;
            unsigned short e8[8];
            *(short8*)e8 = *(const short8*)(ey + o);
            unsigned char w8[8];
            *(unsigned long long*)w8 =
                *(const unsigned long long*)(offs + o);
            if (MODE == POOL_AVG) {
#pragma unroll
              for (int j = 0; j < 8; ++j)
                g[j] += bf2f(e8[j]) / (float)max((int)w8[j], 1);
            } else {
              // does window (oy, ox)'s stored RELATIVE winner code
              // point at THIS input pixel?
              const int rcode = (ih - oy * d.SY) * d.KX +
                                (iw - ox * d.SX);
#pragma unroll
              for (int j = 0; j < 8; ++j)
                if ((int)w8[j] == rcode) g[j] += bf2f(e8[j]);
            }
          }
        unsigned short o8[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) o8[j] = f2bf(g[j]);
        *(short8*)(ex + idx0) = *(short8*)o8;
      }
    } else {
      for (int c = c0; c < d.C; c += C2) {
        const long idx = (long)row * d.C + c;
        float g = 0.f;
        for (int oy = oy0; oy <= oy1; ++oy)
          for (int ox = ox0; ox <= ox1; ++ox) {
            long o = (((long)n * d.OH + oy) * d.OW + ox) * d.C + c;
            if (MODE == POOL_AVG) {
              g += bf2f(ey[o]) / (float)max((int)offs[o], 1);
            } else {
              const int rcode = (ih - oy * d.SY) * d.KX +
                                (iw - ox * d.SX);
              if ((int)offs[o] == rcode) g += bf2f(ey[o]);
            }
          }
        ex[idx] = f2bf(g);
      }
    }
  }
}

// -- backward, specialized for <= WW covering windows per axis (WW = 1
// covers the k2s2 CIFAR pools, WW = 2 the k3s2 AlexNet pools): the
// window set is compile-time unrolled with masked contributions and
// ALL (ey, offs) loads issued together, so their latency overlaps
// instead of serializing per window — the generic kernel measured
// 43.7% SQ_WAIT_INST_ANY (issue-bound) on pool1 (profiles/perf_r2.md).
// vec8 (C % 8 == 0) only; the launcher falls back otherwise.
template <int MODE, int WW>
__global__ void pool_bwd_w_k(const unsigned short* __restrict__ ey,
                             unsigned short* __restrict__ ex,
                             const unsigned char* __restrict__ offs,
                             PoolDesc d, int lg) {
  const int C2 = 1 << lg;
  const int c0 = threadIdx.x & (C2 - 1);
  const int rin = threadIdx.x >> lg;
  const int rpb = blockDim.x >> lg;
  const unsigned rows = (unsigned)d.N * d.H * d.W;
  const int noct = d.C >> 3;
  constexpr int NQ = WW * WW;
  for (unsigned row = blockIdx.x * rpb + rin; row < rows;
       row += gridDim.x * rpb) {
    unsigned t = row / d.W;
    const int iw = row - t * d.W;
    const int n = t / d.H;
    const int ih = t - n * d.H;
    const int oy0 = (ih - d.KY + 1 <= 0) ? 0 : (ih - d.KY) / d.SY + 1;
    const int oy1 = min(d.OH - 1, ih / d.SY);
    const int ox0 = (iw - d.KX + 1 <= 0) ? 0 : (iw - d.KX) / d.SX + 1;
    const int ox1 = min(d.OW - 1, iw / d.SX);
    // candidate windows; out-of-range slots clamp to a valid address
    // (duplicate, L1-hot) and mask their contribution to zero
    int oys[WW], oxs[WW];
    bool vy[WW], vx[WW];
    oys[0] = min(oy0, oy1);
    vy[0] = oy0 <= oy1;
    oxs[0] = min(ox0, ox1);
    vx[0] = ox0 <= ox1;
    if (WW == 2) {
      oys[WW - 1] = min(oys[0] + 1, oy1);
      vy[WW - 1] = vy[0] && oys[0] + 1 <= oy1;
      oxs[WW - 1] = min(oxs[0] + 1, ox1);
      vx[WW - 1] = vx[0] && oxs[0] + 1 <= ox1;
    }
    long ob[NQ];
    int rc[NQ];
    bool vv[NQ];
#pragma unroll
    for (int q = 0; q < NQ; ++q) {
      const int oy = oys[q / WW], ox = oxs[q % WW];
      ob[q] = (((long)n * d.OH + oy) * d.OW + ox) * d.C;
      rc[q] = (ih - oy * d.SY) * d.KX + (iw - ox * d.SX);
      vv[q] = vy[q / WW] && vx[q % WW];
    }
    for (int oc = c0; oc < noct; oc += C2) {
      const int cb = oc * 8;
      unsigned short e8[NQ][8];
      unsigned char w8[NQ][8];
#pragma unroll
      for (int q = 0; q < NQ; ++q) {
        *(short8*)e8[q] = *(const short8*)(ey + ob[q] + cb);
        *(unsigned long long*)w8[q] =
            *(const unsigned long long*)(offs + ob[q] + cb);
      }
      float g[8] = {};
#pragma unroll
      for (int q = 0; q < NQ; ++q) {
        if (MODE == POOL_AVG) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            g[j] += vv[q]
                        ? bf2f(e8[q][j]) / (float)max((int)w8[q][j], 1)
                        : 0.f;
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            g[j] += (vv[q] && (int)w8[q][j] == rc[q]) ? bf2f(e8[q][j])
                                                      : 0.f;
        }
      }
      unsigned short o8[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] = f2bf(g[j]);
      *(short8*)(ex + (long)row * d.C + cb) = *(short8*)o8;
    }
  }
}

void pool_fwd_launch(const unsigned short* x, unsigned short* y,
                     unsigned char* offs,
                     const PoolDesc& d, int mode, hipStream_t s) {
  const int lg = (d.C & 7) == 0 ? chan_lg(d.C >> 3) : chan_lg(d.C);
  const int rpb = 256 >> lg;
  long rows = (long)d.N * d.OH * d.OW;
  dim3 grid((unsigned)min((rows + rpb - 1) / rpb, (long)32768)), block(256);
#define GO(M_)                                                              \
  hipLaunchKernelGGL(pool_fwd_k<M_>, grid, block, 0, s, x, y, offs, d, lg)
  switch (mode) {
    case POOL_AVG: GO(POOL_AVG); break;
    case POOL_MAXABS: GO(POOL_MAXABS); break;
    case POOL_STOCHASTIC: GO(POOL_STOCHASTIC); break;
    default: GO(POOL_MAX); break;
  }
#undef GO
  HIP_CHECK_LAST();
}

void pool_bwd_launch(const unsigned short* ey, unsigned short* ex,
                     const unsigned char* offs, const PoolDesc& d,
                     int mode, hipStream_t s) {
  long rows = (long)d.N * d.H * d.W;
  const int wy = (d.KY - 1) / d.SY + 1;   // covering windows per axis
  const int wx = (d.KX - 1) / d.SX + 1;
  const int noct = d.C >> 3;
  // specialized kernel wins where the round-up lane mapping idles
  // lanes (C = 96: 12 octets over 16 lanes = 25% idle; measured
  // pool1 bwd 113.5 -> 67.1 us); on pow2 octet counts the generic
  // kernel's serial window loop is already ~even (pool2 54.2 vs 56.4)
  if ((d.C & 7) == 0 && wy <= 2 && wx <= 2 &&
      (noct & (noct - 1)) != 0) {
    // full-lane mapping: largest pow2 DIVIDING the octet count, so no
    // lane idles on C = 96-style shapes (12 octets over 16 lanes was
    // 25% idle); the octet loop runs noct/C2 full iterations instead
    int lg = chan_lg(d.C >> 3);
    while (lg > 0 && ((d.C >> 3) & ((1 << lg) - 1)) != 0) --lg;
    const int rpb = 256 >> lg;
    dim3 grid((unsigned)min((rows + rpb - 1) / rpb, (long)32768)),
        block(256);
    const int ww = max(wy, wx);
#define GOW(M_, W_)                                                         \
    hipLaunchKernelGGL((pool_bwd_w_k<M_, W_>), grid, block, 0, s, ey, ex,   \
                       offs, d, lg)
    if (mode == POOL_AVG) {
      if (ww == 1) GOW(POOL_AVG, 1); else GOW(POOL_AVG, 2);
    } else {
      if (ww == 1) GOW(POOL_MAX, 1); else GOW(POOL_MAX, 2);
    }
#undef GOW
    HIP_CHECK_LAST();
    return;
  }
  const int lg = (d.C & 7) == 0 ? chan_lg(d.C >> 3) : chan_lg(d.C);
  const int rpb = 256 >> lg;
  dim3 grid((unsigned)min((rows + rpb - 1) / rpb, (long)32768)), block(256);
#define GO(M_)                                                              \
  hipLaunchKernelGGL(pool_bwd_k<M_>, grid, block, 0, s, ey, ex, offs, d, lg)
  switch (mode) {
    case POOL_AVG: GO(POOL_AVG); break;
    default: GO(POOL_MAX); break;  // max/maxabs/stochastic share offsets
  }
#undef GO
  HIP_CHECK_LAST();
}

// -- cross-channel LRN ------------------------------------------------------
//   y = x * (k + alpha * Σ_{j∈window} x_j²)^(-beta) ; denom saved fp32
// Octet-vectorized: each thread owns 8 consecutive channels — 3 short8
// halo loads, window sums in registers, vector stores.  Requires
// half ≤ 8 (AlexNet win=5 ⇒ half=2).
template <int WIN>
__global__ void lrn_fwd_k(const unsigned short* __restrict__ x,
                          unsigned short* __restrict__ y,
                          float* __restrict__ denom, unsigned rows, int C,
                          int lgo, float alpha, float beta, float kk,
                          bool b34) {
  // OPT adjacent octets per thread-slot: the ±1-octet halo is amortized
  // over OPT output octets ((OPT+2)/OPT load ratio instead of 3×;
  // OPT=2 — OPT=4 measured SLOWER: it cut the thread count to ~1 wave
  // per SIMD slot and the memory-wait-bound kernel lost the
  // concurrency that hides latency), and
  // denom is OPTIONAL — the backward recomputes it from its own x halo,
  // which removes the fp32 round-trip entirely (PMC: both kernels were
  // 65% memory-wait).
  constexpr int OPT = 2;
  const int O2 = 1 << lgo;               // pow2 slot count per row
  const int o0 = threadIdx.x & (O2 - 1);
  const int rin = threadIdx.x >> lgo;
  const int rpb = blockDim.x >> lgo;
  constexpr int half = WIN / 2;
  const int noct = (C + 7) >> 3;
  const int nslot = (noct + OPT - 1) / OPT;
  for (unsigned row = blockIdx.x * rpb + rin; row < rows;
       row += gridDim.x * rpb) {
    const unsigned short* xr = x + (long)row * C;
    float* dr = denom ? denom + (long)row * C : nullptr;
    unsigned short* yr = y + (long)row * C;
    for (int sl = o0; sl < nslot; sl += O2) {
      const int cb = sl * OPT * 8;
      float v[(OPT + 2) * 8];
#pragma unroll
      for (int h = 0; h < OPT + 2; ++h) {
        const int c8 = cb + (h - 1) * 8;
        if (c8 >= 0 && c8 + 8 <= C) {
          unsigned short t8[8];
          *(short8*)t8 = *(const short8*)(xr + c8);
#pragma unroll
          for (int j = 0; j < 8; ++j) v[h * 8 + j] = bf2f(t8[j]);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            int c = c8 + j;
            v[h * 8 + j] = (c >= 0 && c < C) ? bf2f(xr[c]) : 0.f;
          }
        }
      }
#pragma unroll
      for (int t = 0; t < OPT; ++t) {
        const int co = cb + t * 8;
        if (co >= C) break;
        unsigned short o8[8];
        float d8[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float s = 0.f;
#pragma unroll
          for (int w = -half; w <= half; ++w) {
            float tt = v[(t + 1) * 8 + j + w];
            s += tt * tt;
          }
          float d = kk + alpha * s;
          d8[j] = d;
          o8[j] = f2bf(v[(t + 1) * 8 + j] * pow_nbeta(d, beta, b34));
        }
        if (co + 8 <= C) {
          *(short8*)(yr + co) = *(short8*)o8;
          if (dr) {
            *(f32x4*)(dr + co) = *(f32x4*)d8;
            *(f32x4*)(dr + co + 4) = *(f32x4*)(d8 + 4);
          }
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            if (co + j < C) {
              yr[co + j] = o8[j];
              if (dr) dr[co + j] = d8[j];
            }
        }
      }
    }
  }
}

template <int WIN>
__global__ void lrn_bwd_k(const unsigned short* __restrict__ ey,
                          const unsigned short* __restrict__ x,
                          const float* __restrict__ denom,
                          unsigned short* __restrict__ ex, unsigned rows,
                          int C, int lgo, float alpha, float beta,
                          float kk, bool b34) {
  // denom == nullptr: recompute d_j = kk + α·Σ x² in registers — the
  // ±(2·half)-wide x span needed fits inside the same ±1-octet halo
  // for WIN ≤ 9, trading ~60 extra fma per octet for a full fp32
  // tensor read (the kernel is memory-wait bound, the VALU is idle).
  constexpr int OPT = 2;
  const int O2 = 1 << lgo;
  const int o0 = threadIdx.x & (O2 - 1);
  const int rin = threadIdx.x >> lgo;
  const int rpb = blockDim.x >> lgo;
  constexpr int half = WIN / 2;
  static_assert(2 * half <= 8, "halo octet must cover d's x-span");
  const int noct = (C + 7) >> 3;
  const int nslot = (noct + OPT - 1) / OPT;
  for (unsigned row = blockIdx.x * rpb + rin; row < rows;
       row += gridDim.x * rpb) {
    const unsigned short* er = ey + (long)row * C;
    const unsigned short* xr = x + (long)row * C;
    const float* dr = denom ? denom + (long)row * C : nullptr;
    unsigned short* or_ = ex + (long)row * C;
    for (int sl = o0; sl < nslot; sl += O2) {
      const int cb = sl * OPT * 8;
      constexpr int NV = (OPT + 2) * 8;
      float ev[NV], xv[NV], dv[NV];
#pragma unroll
      for (int h = 0; h < OPT + 2; ++h) {
        const int c8 = cb + (h - 1) * 8;
        if (c8 >= 0 && c8 + 8 <= C) {
          unsigned short e8[8], x8[8];
          *(short8*)e8 = *(const short8*)(er + c8);
          *(short8*)x8 = *(const short8*)(xr + c8);
          if (dr) {
            *(f32x4*)(dv + h * 8) = *(const f32x4*)(dr + c8);
            *(f32x4*)(dv + h * 8 + 4) = *(const f32x4*)(dr + c8 + 4);
          }
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            ev[h * 8 + j] = bf2f(e8[j]);
            xv[h * 8 + j] = bf2f(x8[j]);
          }
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            int c = c8 + j;
            bool ok = c >= 0 && c < C;
            ev[h * 8 + j] = ok ? bf2f(er[c]) : 0.f;
            xv[h * 8 + j] = ok ? bf2f(xr[c]) : 0.f;
            if (dr) dv[h * 8 + j] = ok ? dr[c] : 1.f;
          }
        }
      }
      if (!dr) {
        // recompute d over every position whose t_j a window touches:
        // [cb-half, cb+OPT·8+half) ⊂ [cb-8+half, cb+(OPT+1)·8-half)
#pragma unroll
        for (int q = 8 - half; q < NV - 8 + half; ++q) {
          float s = 0.f;
#pragma unroll
          for (int w = -half; w <= half; ++w) {
            float tt = (q + w >= 0 && q + w < NV) ? xv[q + w] : 0.f;
            s += tt * tt;
          }
          dv[q] = kk + alpha * s;
        }
      }
#pragma unroll
      for (int t = 0; t < OPT; ++t) {
        const int co = cb + t * 8;
        if (co >= C) break;
        const int b = (t + 1) * 8;           // center base in v[]
        float tv[8 + 2 * half];
#pragma unroll
        for (int j = 0; j < 8 + 2 * half; ++j) {
          const int q = b - half + j;        // window [co-half, co+7+half]
          tv[j] = ev[q] * xv[q] * pow_nbeta(dv[q], beta, b34) *
                  __builtin_amdgcn_rcpf(dv[q]);
        }
        unsigned short o8[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float acc = 0.f;
#pragma unroll
          for (int w = 0; w < WIN; ++w) acc += tv[j + w];
          float vv = ev[b + j] * pow_nbeta(dv[b + j], beta, b34) -
                     2.f * alpha * beta * xv[b + j] * acc;
          o8[j] = f2bf(vv);
        }
        if (co + 8 <= C) {
          *(short8*)(or_ + co) = *(short8*)o8;
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            if (co + j < C) or_[co + j] = o8[j];
        }
      }
    }
  }
}

// window sizes are small odd ints; compile the common ones (register-
// resident halos need compile-time WIN)
#define LRN_WIN_SWITCH(win, F)                                              \
  switch (win) {                                                            \
    case 3: F(3); break;                                                    \
    case 5: F(5); break;                                                    \
    case 7: F(7); break;                                                    \
    case 9: F(9); break;                                                    \
    default: throw std::runtime_error("LRN win must be 3/5/7/9");           \
  }

static inline int lrn_lgo(int C) {
  const int nslot = (((C + 7) >> 3) + 1) / 2;   // OPT=2 octets per slot
  int lg = 0;
  while ((1 << lg) < nslot && lg < 5) ++lg;
  return lg;
}

void lrn_fwd_launch(const unsigned short* x, unsigned short* y,
                    float* denom, long rows, int C, float alpha,
                    float beta, float k, int win, hipStream_t s) {
  const int lgo = lrn_lgo(C);
  const int rpb = 256 >> lgo;
  dim3 grid((unsigned)min((rows + rpb - 1) / rpb, (long)131072)),
      block(256);
  const bool b34 = beta == 0.75f;
#define GO(W)                                                               \
  hipLaunchKernelGGL(lrn_fwd_k<W>, grid, block, 0, s, x, y, denom,          \
                     (unsigned)rows, C, lgo, alpha, beta, k, b34)
  LRN_WIN_SWITCH(win, GO);
#undef GO
  HIP_CHECK_LAST();
}

void lrn_bwd_launch(const unsigned short* ey, const unsigned short* x,
                    const float* denom, float* ws, unsigned short* ex,
                    long rows, int C, float alpha, float beta, float kk,
                    int win, hipStream_t s) {
  (void)ws;                       // fused kernel needs no workspace
  const bool b34 = beta == 0.75f;
  const int lgo = lrn_lgo(C);
  const int rpb = 256 >> lgo;
  dim3 grid((unsigned)min((rows + rpb - 1) / rpb, (long)131072));
#define GO(W)                                                               \
  hipLaunchKernelGGL(lrn_bwd_k<W>, grid, dim3(256), 0, s, ey, x, denom,     \
                     ex, (unsigned)rows, C, lgo, alpha, beta, kk, b34)
  LRN_WIN_SWITCH(win, GO);
#undef GO
  HIP_CHECK_LAST();
}
