// Pooling + cross-channel LRN kernels (NHWC bf16).
// Replaces reference znicz/ocl/{pooling.cl, gd_pooling.cl,
// normalization.cl} — redesigned: gather-based backward (no atomics),
// c-innermost layout so consecutive lanes stream consecutive channels.
#include "common.h"

// -- forward: thread per output element ------------------------------------
template <int MODE>
__global__ void pool_fwd_k(const unsigned short* __restrict__ x,
                           unsigned short* __restrict__ y,
                           int* __restrict__ offs, PoolDesc d) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)d.N * d.OH * d.OW * d.C;
  if (idx >= total) return;
  int c = idx % d.C;
  long t = idx / d.C;
  int ow = t % d.OW;
  t /= d.OW;
  int oh = t % d.OH;
  int n = t / d.OH;
  const int ih0 = oh * d.SY, iw0 = ow * d.SX;
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
      float v = bf2f(x[off]);
      ++count;
      if (MODE == POOL_AVG) {
        sum += v;
      } else if (MODE == POOL_MAX) {
        if (bestoff < 0 || v > best) { best = v; bestoff = (int)off; }
      } else if (MODE == POOL_MAXABS) {
        float a = fabsf(v);
        if (bestoff < 0 || a > besta) { best = v; besta = a;
                                        bestoff = (int)off; }
      } else {  // stochastic: first cell where running pos-sum crosses
        float pv = fmaxf(v, 0.f);
        stoch_acc += pv;
        if (bestoff < 0 && (stoch_acc > stoch_target && pv > 0.f)) {
          best = v;
          bestoff = (int)off;
        }
      }
    }
  }
  if (MODE == POOL_AVG) {
    y[idx] = f2bf(sum / max(count, 1));
    if (offs) offs[idx] = count;      // store divisor for backward
  } else {
    if (MODE == POOL_STOCHASTIC && bestoff < 0) {
      // all-nonpositive window: fall back to max
      for (int ky = 0; ky < d.KY; ++ky) {
        int ih = ih0 + ky;
        if (ih >= d.H) break;
        for (int kx = 0; kx < d.KX; ++kx) {
          int iw = iw0 + kx;
          if (iw >= d.W) break;
          long off = (((long)n * d.H + ih) * d.W + iw) * d.C + c;
          float v = bf2f(x[off]);
          if (bestoff < 0 || v > best) { best = v; bestoff = (int)off; }
        }
      }
    }
    y[idx] = f2bf(best);
    if (offs) offs[idx] = bestoff;
  }
}

// -- backward: thread per INPUT element (gather, no atomics) ----------------
template <int MODE>
__global__ void pool_bwd_k(const unsigned short* __restrict__ ey,
                           unsigned short* __restrict__ ex,
                           const int* __restrict__ offs, PoolDesc d) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)d.N * d.H * d.W * d.C;
  if (idx >= total) return;
  int c = idx % d.C;
  long t = idx / d.C;
  int iw = t % d.W;
  t /= d.W;
  int ih = t % d.H;
  int n = t / d.H;
  float g = 0.f;
  // output cells whose window covers (ih, iw)
  int oy0 = (ih - d.KY + 1 <= 0) ? 0 : (ih - d.KY) / d.SY + 1;
  int oy1 = min(d.OH - 1, ih / d.SY);
  int ox0 = (iw - d.KX + 1 <= 0) ? 0 : (iw - d.KX) / d.SX + 1;
  int ox1 = min(d.OW - 1, iw / d.SX);
  for (int oy = oy0; oy <= oy1; ++oy)
    for (int ox = ox0; ox <= ox1; ++ox) {
      long o = (((long)n * d.OH + oy) * d.OW + ox) * d.C + c;
      if (MODE == POOL_AVG) {
        g += bf2f(ey[o]) / (float)max(offs[o], 1);
      } else {
        if (offs[o] == (int)idx) g += bf2f(ey[o]);
      }
    }
  ex[idx] = f2bf(g);
}

void pool_fwd_launch(const unsigned short* x, unsigned short* y, int* offs,
                     const PoolDesc& d, int mode, hipStream_t s) {
  long total = (long)d.N * d.OH * d.OW * d.C;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
#define GO(M_)                                                              \
  hipLaunchKernelGGL(pool_fwd_k<M_>, grid, block, 0, s, x, y, offs, d)
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
                     const int* offs, const PoolDesc& d, int mode,
                     hipStream_t s) {
  long total = (long)d.N * d.H * d.W * d.C;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
#define GO(M_)                                                              \
  hipLaunchKernelGGL(pool_bwd_k<M_>, grid, block, 0, s, ey, ex, offs, d)
  switch (mode) {
    case POOL_AVG: GO(POOL_AVG); break;
    default: GO(POOL_MAX); break;  // max/maxabs/stochastic share offsets
  }
#undef GO
  HIP_CHECK_LAST();
}

// -- cross-channel LRN ------------------------------------------------------
//   y = x * (k + alpha * Σ_{j∈window} x_j²)^(-beta) ; denom saved fp32
// Thread per ELEMENT (c innermost ⇒ lane-coalesced; window neighbours
// are L1 hits), one fast powf each — not per (c, j) pair.
__global__ void lrn_fwd_k(const unsigned short* __restrict__ x,
                          unsigned short* __restrict__ y,
                          float* __restrict__ denom, long total, int C,
                          float alpha, float beta, float kk, int win) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const int half = win / 2;
  for (long p = i; p < total; p += stride) {
    int c = (int)(p % C);
    long r0 = p - c;
    int j0 = max(0, c - half), j1 = min(C - 1, c + half);
    float s = 0.f;
    for (int j = j0; j <= j1; ++j) {
      float v = bf2f(x[r0 + j]);
      s += v * v;
    }
    float d = kk + alpha * s;
    denom[p] = d;
    y[p] = f2bf(bf2f(x[p]) * __powf(d, -beta));
  }
}

// backward in two fully-parallel passes (ws = fp32 workspace, numel total):
//   pass 1:  t_j = dy_j·x_j·D_j^{-β-1}
__global__ void lrn_bwd_pre_k(const unsigned short* __restrict__ ey,
                              const unsigned short* __restrict__ x,
                              const float* __restrict__ denom,
                              float* __restrict__ t, long total,
                              float beta) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < total; p += stride)
    t[p] = bf2f(ey[p]) * bf2f(x[p]) * __powf(denom[p], -beta - 1.f);
}

//   pass 2:  dx_c = dy_c·D_c^{-β} − 2αβ·x_c · Σ_{|j−c|≤half} t_j
__global__ void lrn_bwd_k(const unsigned short* __restrict__ ey,
                          const unsigned short* __restrict__ x,
                          const float* __restrict__ denom,
                          const float* __restrict__ t,
                          unsigned short* __restrict__ ex, long total,
                          int C, float alpha, float beta, int win) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const int half = win / 2;
  for (long p = i; p < total; p += stride) {
    int c = (int)(p % C);
    long r0 = p - c;
    int j0 = max(0, c - half), j1 = min(C - 1, c + half);
    float acc = 0.f;
    for (int j = j0; j <= j1; ++j) acc += t[r0 + j];
    float v = bf2f(ey[p]) * __powf(denom[p], -beta) -
              2.f * alpha * beta * bf2f(x[p]) * acc;
    ex[p] = f2bf(v);
  }
}

void lrn_fwd_launch(const unsigned short* x, unsigned short* y,
                    float* denom, long rows, int C, float alpha,
                    float beta, float k, int win, hipStream_t s) {
  long total = rows * C;
  int blocks = (int)min((total + 255) / 256, (long)8192);
  hipLaunchKernelGGL(lrn_fwd_k, dim3(blocks), dim3(256), 0, s, x, y,
                     denom, total, C, alpha, beta, k, win);
  HIP_CHECK_LAST();
}

void lrn_bwd_launch(const unsigned short* ey, const unsigned short* x,
                    const float* denom, float* ws, unsigned short* ex,
                    long rows, int C, float alpha, float beta, int win,
                    hipStream_t s) {
  long total = rows * C;
  int blocks = (int)min((total + 255) / 256, (long)8192);
  hipLaunchKernelGGL(lrn_bwd_pre_k, dim3(blocks), dim3(256), 0, s, ey, x,
                     denom, ws, total, beta);
  HIP_CHECK_LAST();
  hipLaunchKernelGGL(lrn_bwd_k, dim3(blocks), dim3(256), 0, s, ey, x,
                     denom, ws, ex, total, C, alpha, beta, win);
  HIP_CHECK_LAST();
}
