// Elementwise CDNA4 kernels: activations, dropout (counter-based PRNG),
// fused SGD apply (fp32 master + bf16 refresh), column reduce, transpose.
// Replaces reference znicz/ocl/{err_output.cl, dropout.cl,
// gradient_descent.cl weights/bias_update, random.cl} — redesigned:
// vectorized bf16x8 loads (wave64 coalescing), no PRNG state buffers,
// SGD applied once per (all-reduced) gradient bucket.
#include "common.h"

// -- activation forward/backward (in-place on bf16 tensors) -----------------

template <int A>
__global__ void act_fwd_k(unsigned short* x, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i * 8; p < n; p += stride * 8) {
    unsigned short v[8];
    const long todo = min((long)8, n - p);
    if (todo == 8) {
      *(short8*)v = *(const short8*)(x + p);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = f2bf(act_fwd<A>(bf2f(v[j])));
      *(short8*)(x + p) = *(const short8*)v;
    } else {
      for (int j = 0; j < todo; ++j)
        x[p + j] = f2bf(act_fwd<A>(bf2f(x[p + j])));
    }
  }
}

// err ∘= act'(y)
template <int A>
__global__ void act_bwd_k(unsigned short* err, const unsigned short* y,
                          long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i * 8; p < n; p += stride * 8) {
    const long todo = min((long)8, n - p);
    if (todo == 8) {
      unsigned short e[8], v[8];
      *(short8*)e = *(const short8*)(err + p);
      *(short8*)v = *(const short8*)(y + p);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        e[j] = f2bf(bf2f(e[j]) * act_dy<A>(bf2f(v[j])));
      *(short8*)(err + p) = *(const short8*)e;
    } else {
      for (int j = 0; j < todo; ++j)
        err[p + j] = f2bf(bf2f(err[p + j]) * act_dy<A>(bf2f(y[p + j])));
    }
  }
}

void act_fwd_launch(unsigned short* x, long n, int act, hipStream_t s) {
  int blocks = (int)min((n + 8 * 256 - 1) / (8 * 256), (long)2048);
#define GO(A) hipLaunchKernelGGL(act_fwd_k<A>, dim3(blocks), dim3(256), 0, \
                                 s, x, n)
  ACT_SWITCH(act, GO);
#undef GO
  HIP_CHECK_LAST();
}

void act_bwd_launch(unsigned short* err, const unsigned short* y, long n,
                    int act, hipStream_t s) {
  int blocks = (int)min((n + 8 * 256 - 1) / (8 * 256), (long)2048);
#define GO(A) hipLaunchKernelGGL(act_bwd_k<A>, dim3(blocks), dim3(256), 0, \
                                 s, err, y, n)
  ACT_SWITCH(act, GO);
#undef GO
  HIP_CHECK_LAST();
}

// -- fused SGD: master fp32 update + bf16 working-copy refresh --------------
//  g = grad*scale + decay*(l1*sign(w) + (1-l1)*w)
//  v = moment*v + lr*g ;  w -= v ;  bf16[i] = w
#include "sgd_multi.h"

__device__ __forceinline__ void sgd_elem(
    float* __restrict__ w, const float* __restrict__ g,
    float* __restrict__ vel, unsigned short* __restrict__ wb, long p,
    float lr, float decay, float l1, float moment, float scale) {
  float wi = w[p];
  float l1l2 = l1 * (wi > 0.f ? 1.f : (wi < 0.f ? -1.f : 0.f)) +
               (1.f - l1) * wi;
  float gi = g[p] * scale + decay * l1l2;
  float v = moment * vel[p] + lr * gi;
  if (moment != 0.f) vel[p] = v;
  wi -= v;
  w[p] = wi;
  if (wb) wb[p] = f2bf(wi);
}

__global__ void sgd_k(float* __restrict__ w, const float* __restrict__ g,
                      float* __restrict__ vel,
                      unsigned short* __restrict__ wb, long n, float lr,
                      float decay, float l1, float moment, float scale) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride)
    sgd_elem(w, g, vel, wb, p, lr, decay, l1, moment, scale);
}

// one launch for a whole gradient bucket: block -> tensor via b2t.
// Each thread owns 8 CONSECUTIVE elements (2× float4 per stream) — the
// strided-single-float version measured 86% memory-wait at 4.6 TB/s.
__global__ void sgd_multi_k(const SgdDesc* __restrict__ ds,
                            const int* __restrict__ b2t) {
  const SgdDesc d = ds[b2t[blockIdx.x]];
  const long base = ((long)blockIdx.x - d.blk0) * SGD_MULTI_SPAN;
  const long p0 = base + (long)threadIdx.x * 8;
  if (p0 + 8 <= d.n) {
    f32x4 w[2], g[2], vel[2];
    w[0] = *(const f32x4*)(d.w + p0);
    w[1] = *(const f32x4*)(d.w + p0 + 4);
    g[0] = *(const f32x4*)(d.g + p0);
    g[1] = *(const f32x4*)(d.g + p0 + 4);
    vel[0] = *(const f32x4*)(d.vel + p0);
    vel[1] = *(const f32x4*)(d.vel + p0 + 4);
    unsigned short wb8[8];
#pragma unroll
    for (int h = 0; h < 2; ++h)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float wi = w[h][j];
        float l1l2 = d.l1 * (wi > 0.f ? 1.f : (wi < 0.f ? -1.f : 0.f)) +
                     (1.f - d.l1) * wi;
        float gi = g[h][j] * d.scale + d.decay * l1l2;
        float v = d.moment * vel[h][j] + d.lr * gi;
        vel[h][j] = v;
        wi -= v;
        w[h][j] = wi;
        wb8[h * 4 + j] = f2bf(wi);
      }
    *(f32x4*)(d.w + p0) = w[0];
    *(f32x4*)(d.w + p0 + 4) = w[1];
    if (d.moment != 0.f) {
      *(f32x4*)(d.vel + p0) = vel[0];
      *(f32x4*)(d.vel + p0 + 4) = vel[1];
    }
    if (d.wb) *(short8*)(d.wb + p0) = *(short8*)wb8;
  } else {
    for (long p = p0; p < d.n; ++p)
      sgd_elem(d.w, d.g, d.vel, d.wb, p, d.lr, d.decay, d.l1, d.moment,
               d.scale);
  }
}

void sgd_multi_launch(const void* descs, const int* b2t, int nblocks,
                      hipStream_t s) {
  hipLaunchKernelGGL(sgd_multi_k, dim3(nblocks), dim3(256), 0, s,
                     (const SgdDesc*)descs, b2t);
  HIP_CHECK_LAST();
}

void sgd_launch(float* w, const float* g, float* vel, unsigned short* wb,
                long n, float lr, float decay, float l1, float moment,
                float scale, hipStream_t s) {
  int blocks = (int)min((n + 255) / 256, (long)2048);
  hipLaunchKernelGGL(sgd_k, dim3(blocks), dim3(256), 0, s, w, g, vel, wb,
                     n, lr, decay, l1, moment, scale);
  HIP_CHECK_LAST();
}

// zero-fill on the COMPUTE stream (hipMemsetAsync would bounce to the
// SDMA engine — the cross-engine sync costs ~100 µs per call)
__global__ void zero_f32_k(float* __restrict__ p, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long q = i; q < n; q += stride) p[q] = 0.f;
}

void zero_f32_launch(float* p, long n, hipStream_t s) {
  int b = (int)min((n + 255) / 256, (long)2048);
  hipLaunchKernelGGL(zero_f32_k, dim3(b), dim3(256), 0, s, p, n);
  HIP_CHECK_LAST();
}

// conv dgrad-image refresh: [K, p, C] -> [C, p, K] (p = KH·KW plane
// index).  LDS-tiled batched transpose — replaces a torch strided
// copy_ whose 2-B scattered reads ran at ~1 TB/s.
#define PTT 32
__global__ void permute_kc_k(const unsigned short* __restrict__ src,
                             unsigned short* __restrict__ dst, int K,
                             int P, int C) {
  __shared__ unsigned short t[PTT][PTT + 1];
  const int p = blockIdx.z;
  const int k0 = blockIdx.x * PTT, c0 = blockIdx.y * PTT;
  const int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;   // 32 x 8
#pragma unroll
  for (int dy = 0; dy < PTT; dy += 8) {
    const int k = k0 + ty + dy, c = c0 + tx;
    if (k < K && c < C)
      t[ty + dy][tx] = src[((long)k * P + p) * C + c];
  }
  __syncthreads();
#pragma unroll
  for (int dy = 0; dy < PTT; dy += 8) {
    const int c = c0 + ty + dy, k = k0 + tx;
    if (c < C && k < K)
      dst[((long)c * P + p) * K + k] = t[tx][ty + dy];
  }
}

void permute_kc_launch(const unsigned short* src, unsigned short* dst,
                       int K, int P, int C, hipStream_t s) {
  dim3 grid((K + PTT - 1) / PTT, (C + PTT - 1) / PTT, P);
  hipLaunchKernelGGL(permute_kc_k, grid, dim3(256), 0, s, src, dst, K, P,
                     C);
  HIP_CHECK_LAST();
}

// zero two disjoint buffers in one launch (split-K C workspace + the
// fused bias-gradient accumulator)
__global__ void zero2_f32_k(float* __restrict__ p, long n,
                            float* __restrict__ q, long m) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = i; t < n; t += stride) p[t] = 0.f;
  for (long t = i; t < m; t += stride) q[t] = 0.f;
}

void zero2_f32_launch(float* p, long n, float* q, long m, hipStream_t s) {
  int b = (int)min((n + 255) / 256, (long)2048);
  hipLaunchKernelGGL(zero2_f32_k, dim3(b), dim3(256), 0, s, p, n, q, m);
  HIP_CHECK_LAST();
}

// -- column sum (bias gradient): out[N] = Σ_m A[M][N], bf16 -> fp32 --------
// Thread owns 8 consecutive cols (one b128 load per row — wave reads
// 1 KiB contiguous); row-slab blocks reduce through LDS, then one fp32
// atomic per (octet, slab).  Fills 256 CUs on tall-skinny conv bias
// shapes (M ≈ 10⁵..10⁶, N ≤ a few hundred).
__global__ void col_sum_k(const unsigned short* __restrict__ a,
                          float* __restrict__ ws, int M, int N,
                          int rows_per_slab, int lgo, int accumulate) {
  const int O2 = 1 << lgo;
  const int oct = threadIdx.x & (O2 - 1);
  const int rsub = threadIdx.x >> lgo;
  const int rstep = blockDim.x >> lgo;
  const int cb = oct * 8;
  float s[8] = {};
  const long m0 = (long)blockIdx.x * rows_per_slab;
  const long m1 = min((long)M, m0 + rows_per_slab);
  if (cb < N) {
    if (cb + 8 <= N && (N & 7) == 0) {
      const unsigned short* p = a + (m0 + rsub) * (long)N + cb;
      for (long m = m0 + rsub; m < m1; m += rstep,
                p += (long)N * rstep) {
        unsigned short v[8];
        *(short8*)v = *(const short8*)p;
#pragma unroll
        for (int j = 0; j < 8; ++j) s[j] += bf2f(v[j]);
      }
    } else {
      for (long m = m0 + rsub; m < m1; m += rstep) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (cb + j < N) s[j] += bf2f(a[m * (long)N + cb + j]);
      }
    }
  }
  __shared__ float red[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x][j] = s[j];
  __syncthreads();
  if (rsub == 0 && cb < N) {
#pragma unroll 1
    for (int r = 1; r < rstep; ++r)
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += red[oct + r * O2][j];
    // plain stores into the per-slab workspace row (NO atomics: 10^3
    // blocks x few cache lines of fp32 fadds serialize at the L2 slice
    // owning them -- measured as a flat ~190 us floor)
    float* dst = ws + (long)blockIdx.x * N + cb;  // ws==out at 1 slab
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (cb + j < N)
        dst[j] = (accumulate && gridDim.x == 1 ? dst[j] : 0.f) + s[j];
  }
}

// pass 2: out[col] = sum_s ws[s][col] — [slabs, N] fp32.  One block per
// 8-column octet; threads stripe the slab dim (32 independent loads in
// flight per lane), LDS tree-reduce.  A naive col-per-thread loop here
// serialized on vmcnt and cost ~0.2 µs PER SLAB (the measured flat
// ~200 µs floor of the whole col_sum) — see tools/probe_colsum.hip.
__global__ void col_sum_fin_k(const float* __restrict__ ws,
                              float* __restrict__ out, int slabs, int N,
                              int accumulate) {
  const int cb = blockIdx.x * 8;
  const int j = threadIdx.x & 7;        // column within octet
  const int r0 = threadIdx.x >> 3;      // slab stripe (32 lanes)
  const int col = cb + j;
  float s = 0.f;
  if (col < N) {
    for (int r = r0; r < slabs; r += 32)
      s += ws[(long)r * N + col];
  }
  __shared__ float red[256];
  red[threadIdx.x] = s;
  __syncthreads();
#pragma unroll
  for (int step = 128; step >= 8; step >>= 1) {
    if (threadIdx.x < step) red[threadIdx.x] += red[threadIdx.x + step];
    __syncthreads();
  }
  if (threadIdx.x < 8 && col < N)
    out[col] = (accumulate ? out[col] : 0.f) + red[threadIdx.x];
}

// wide-N fallback (N ≥ 2048 exceeds one block's octet span; those
// matrices are short FC-bias shapes where column parallelism suffices)
__global__ void col_sum_wide_k(const unsigned short* __restrict__ a,
                               float* __restrict__ out, int M, int N,
                               int rows_per_slab, int accumulate) {
  int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= N) return;
  long m0 = (long)blockIdx.y * rows_per_slab;
  long m1 = min((long)M, m0 + rows_per_slab);
  float s = 0.f;
  const unsigned short* p = a + m0 * N + col;
  for (long m = m0; m < m1; ++m, p += N) s += bf2f(*p);
  if (gridDim.y == 1)
    out[col] = (accumulate ? out[col] : 0.f) + s;
  else
    unsafeAtomicAdd(&out[col], s);
}

void col_sum_launch(const unsigned short* a, float* out, int M, int N,
                    bool accumulate, hipStream_t s) {
  if (N >= 2048) {
    int bx = (N + 255) / 256;
    int slabs = (int)min((long)(M + 63) / 64, (long)max(1, 1024 / bx));
    if (slabs > 1 && !accumulate) zero_f32_launch(out, N, s);
    int rps = (M + slabs - 1) / slabs;
    hipLaunchKernelGGL(col_sum_wide_k, dim3(bx, slabs), dim3(256), 0, s,
                       a, out, M, N, rps, (int)accumulate);
    HIP_CHECK_LAST();
    return;
  }
  const int noct = (N + 7) >> 3;
  int lgo = 0;
  while ((1 << lgo) < noct && lgo < 8) ++lgo;
  int slabs = (int)min((long)(M + 63) / 64, (long)512);
  if (slabs == 1) {
    hipLaunchKernelGGL(col_sum_k, dim3(1), dim3(256), 0, s, a,
                       out, M, N, M, lgo, (int)accumulate);
    HIP_CHECK_LAST();
    return;
  }
  // persistent per-device workspace for the slab partials (<= 8 MiB)
  static float* ws = nullptr;
  static size_t ws_cap = 0;
  size_t need = (size_t)slabs * N * sizeof(float);
  if (need > ws_cap) {
    if (ws) (void)hipFree(ws);
    ws_cap = need < (8u << 20) ? (8u << 20) : need;
    if (hipMalloc(&ws, ws_cap) != hipSuccess)
      throw std::runtime_error("col_sum workspace alloc failed");
  }
  int rps = (M + slabs - 1) / slabs;
  hipLaunchKernelGGL(col_sum_k, dim3(slabs), dim3(256), 0, s, a,
                     ws, M, N, rps, lgo, 0);
  HIP_CHECK_LAST();
  hipLaunchKernelGGL(col_sum_fin_k, dim3((N + 7) / 8), dim3(256), 0,
                     s, ws, out, slabs, N, (int)accumulate);
  HIP_CHECK_LAST();
}

// -- bf16 2-D transpose (LDS 32x32 tiles, +1 pad) ---------------------------
// 64×64 tiles, 16-B vector global access on BOTH sides: each thread
// vector-loads one short8 chunk, scatters its 8 elements to TRANSPOSED
// LDS positions (the scalar cost sits on the cheap LDS-write side),
// then vector-reads the transposed image and stores coalesced.  The
// old 32×32 scalar-u16 version ran ~2.5 TB/s (2-B store grain).
__global__ void transpose2d_k(const unsigned short* __restrict__ in,
                              unsigned short* __restrict__ out, int R,
                              int C) {
  constexpr int TP = 64 + 8;             // pitch: 16-B aligned, detunes
  __shared__ unsigned short tt[64 * TP]; // banks across the 8 writes
  const int c0 = blockIdx.x * 64, r0 = blockIdx.y * 64;
  const int tx = threadIdx.x & 7;        // chunk (8 cols) within row
  const int ty = threadIdx.x >> 3;       // row 0..31; two row-halves
  const bool interior = (r0 + 64 <= R) && (c0 + 64 <= C);
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int r = ty + h * 32;           // tile row
    const int c = tx * 8;                // tile col of this chunk
    unsigned short v[8];
    if (interior) {
      *(short8*)v = *(const short8*)(in + (long)(r0 + r) * C + c0 + c);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int rr = r0 + r, cc = c0 + c + j;
        v[j] = (rr < R && cc < C) ? in[(long)rr * C + cc]
                                  : (unsigned short)0;
      }
    }
    // row index XORed with the column-octet spreads the 8 scatter
    // writes of a wave over 32 distinct banks (unswizzled, every lane
    // of an instruction hits the same 4-bank set: 16-way conflict)
    const int swz = (tx & 7) << 3;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      tt[(c + j) * TP + (r ^ swz)] = v[j];
  }
  __syncthreads();
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int tr = ty + h * 32;          // transposed-tile row (= col c)
    const int tc = tx * 8;
    const int swz = ((tr >> 3) & 7) << 3;
    short8 v = *(const short8*)(tt + tr * TP + (tc ^ swz));
    const int orow = c0 + tr, ocol = r0 + tc;
    if (interior) {
      *(short8*)(out + (long)orow * R + ocol) = v;
    } else if (orow < C) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (ocol + j < R)
          out[(long)orow * R + ocol + j] = ((unsigned short*)&v)[j];
    }
  }
}

void transpose2d_launch(const unsigned short* in, unsigned short* out,
                        int R, int C, hipStream_t s) {
  dim3 grid((C + 63) / 64, (R + 63) / 64);
  hipLaunchKernelGGL(transpose2d_k, grid, dim3(256), 0, s, in, out, R, C);
  HIP_CHECK_LAST();
}

// -- misc small kernels for cold units (depooling, RBM, Kohonen, input
// normalization, Uniform) ---------------------------------------------------

// dst[i] = src[offs[i]]  (depooling backward / unpooling gather)
__global__ void gather_off_k(const unsigned short* __restrict__ src,
                             const int* __restrict__ offs,
                             unsigned short* __restrict__ dst, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride) {
    int o = offs[p];
    dst[p] = o >= 0 ? src[o] : (unsigned short)0;
  }
}

void gather_off_launch(const unsigned short* src, const int* offs,
                       unsigned short* dst, long n, hipStream_t s) {
  int b = (int)min((n + 255) / 256, (long)4096);
  hipLaunchKernelGGL(gather_off_k, dim3(b), dim3(256), 0, s, src, offs,
                     dst, n);
  HIP_CHECK_LAST();
}

// dst[offs[i]] = src[i], dst pre-zeroed (depooling forward scatter; max-
// pool offsets are unique per output cell so no atomics are needed)
__global__ void scatter_off_k(const unsigned short* __restrict__ src,
                              const int* __restrict__ offs,
                              unsigned short* __restrict__ dst, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride) {
    int o = offs[p];
    if (o >= 0) dst[o] = src[p];
  }
}

__global__ void zero_bf16_k(unsigned short* __restrict__ p, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long q = i; q < n; q += stride) p[q] = 0;
}

void scatter_off_launch(const unsigned short* src, const int* offs,
                        unsigned short* dst, long n, long n_dst,
                        hipStream_t s) {
  int bz = (int)min((n_dst + 255) / 256, (long)4096);
  hipLaunchKernelGGL(zero_bf16_k, dim3(bz), dim3(256), 0, s, dst, n_dst);
  int b = (int)min((n + 255) / 256, (long)4096);
  hipLaunchKernelGGL(scatter_off_k, dim3(b), dim3(256), 0, s, src, offs,
                     dst, n);
  HIP_CHECK_LAST();
}

// y = U(seed, i) < x  (Bernoulli sampling for RBM binarization)
__global__ void bernoulli_k(const unsigned short* __restrict__ x,
                            unsigned short* __restrict__ y, long n,
                            unsigned long long seed) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride)
    y[p] = uniform01(seed, (unsigned long long)p) < bf2f(x[p])
               ? (unsigned short)0x3F80   // bf16 1.0
               : (unsigned short)0;
}

void bernoulli_launch(const unsigned short* x, unsigned short* y, long n,
                      unsigned long long seed, hipStream_t s) {
  int b = (int)min((n + 255) / 256, (long)4096);
  hipLaunchKernelGGL(bernoulli_k, dim3(b), dim3(256), 0, s, x, y, n, seed);
  HIP_CHECK_LAST();
}

// uniform fill in [lo, hi) (the reference's Uniform accelerated unit)
__global__ void fill_uniform_k(float* __restrict__ y, long n, float lo,
                               float hi, unsigned long long seed) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride)
    y[p] = lo + (hi - lo) * uniform01(seed, (unsigned long long)p);
}

void fill_uniform_launch(float* y, long n, float lo, float hi,
                         unsigned long long seed, hipStream_t s) {
  int b = (int)min((n + 255) / 256, (long)4096);
  hipLaunchKernelGGL(fill_uniform_k, dim3(b), dim3(256), 0, s, y, n, lo,
                     hi, seed);
  HIP_CHECK_LAST();
}

// per-sample argmin over fp32 rows (Kohonen winner search): one wave per
// row, lane-strided min + DPP-free shuffle reduce via LDS
__global__ void row_argmin_k(const float* __restrict__ d,
                             int* __restrict__ winners, int rows,
                             int cols) {
  __shared__ float bv[256];
  __shared__ int bi[256];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const float* dr = d + (long)row * cols;
  float best = 3.4e38f;
  int besti = 0;
  for (int c = threadIdx.x; c < cols; c += blockDim.x) {
    float v = dr[c];
    if (v < best) { best = v; besti = c; }
  }
  bv[threadIdx.x] = best;
  bi[threadIdx.x] = besti;
  __syncthreads();
  for (int step = blockDim.x / 2; step > 0; step >>= 1) {
    if (threadIdx.x < step) {
      if (bv[threadIdx.x + step] < bv[threadIdx.x] ||
          (bv[threadIdx.x + step] == bv[threadIdx.x] &&
           bi[threadIdx.x + step] < bi[threadIdx.x])) {
        bv[threadIdx.x] = bv[threadIdx.x + step];
        bi[threadIdx.x] = bi[threadIdx.x + step];
      }
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) winners[row] = bi[0];
}

void row_argmin_launch(const float* d, int* winners, int rows, int cols,
                       hipStream_t s) {
  hipLaunchKernelGGL(row_argmin_k, dim3(rows), dim3(256), 0, s, d,
                     winners, rows, cols);
  HIP_CHECK_LAST();
}

// y = (x - mean) * rdisp, mean/rdisp broadcast over the batch dim
// (MeanDispNormalizer; S = sample size)
__global__ void mean_disp_k(const unsigned short* __restrict__ x,
                            const float* __restrict__ mean,
                            const float* __restrict__ rdisp,
                            unsigned short* __restrict__ y, long total,
                            long S) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < total; p += stride) {
    long sidx = p % S;
    y[p] = f2bf((bf2f(x[p]) - mean[sidx]) * rdisp[sidx]);
  }
}

void mean_disp_launch(const unsigned short* x, const float* mean,
                      const float* rdisp, unsigned short* y, long total,
                      long S, hipStream_t s) {
  int b = (int)min((total + 255) / 256, (long)4096);
  hipLaunchKernelGGL(mean_disp_k, dim3(b), dim3(256), 0, s, x, mean,
                     rdisp, y, total, S);
  HIP_CHECK_LAST();
}

// confusion-matrix accumulation: cm[pred, label] += 1, device-side so
// the hot loop never syncs (int32 atomics on a tiny [C,C] matrix)
__global__ void confusion_k(const int* __restrict__ max_idx,
                            const int* __restrict__ labels,
                            int* __restrict__ cm, int batch, int C) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= batch) return;
  int lbl = labels[i];
  if (lbl < 0 || lbl >= C) return;
  int pred = max_idx[i];
  atomicAdd(&cm[pred * C + lbl], 1);
}

void confusion_launch(const int* max_idx, const int* labels, int* cm,
                      int batch, int C, hipStream_t s) {
  hipLaunchKernelGGL(confusion_k, dim3((batch + 255) / 256), dim3(256),
                     0, s, max_idx, labels, cm, batch, C);
  HIP_CHECK_LAST();
}

// -- dropout ---------------------------------------------------------------
// step counter lives in DEVICE memory so the dropout launch sequence is
// hipGraph-replayable: the bump kernel precedes the mask kernel on the
// same stream, so every replay draws a fresh mask (a by-value counter
// would freeze inside a captured graph).
__global__ void ctr_bump_k(unsigned long long* __restrict__ c) {
  if (threadIdx.x == 0 && blockIdx.x == 0) ++*c;
}

__global__ void dropout_fwd_k(const unsigned short* __restrict__ x,
                              unsigned short* __restrict__ y,
                              unsigned char* __restrict__ mask, long n,
                              float ratio, float scale,
                              unsigned long long seed,
                              const unsigned long long* __restrict__ ctr) {
  if (ctr) seed += *ctr;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride) {
    unsigned char keep = uniform01(seed, (unsigned long long)p) >= ratio;
    mask[p] = keep;
    y[p] = keep ? f2bf(bf2f(x[p]) * scale) : (unsigned short)0;
  }
}

__global__ void dropout_bwd_k(unsigned short* __restrict__ err,
                              const unsigned char* __restrict__ mask,
                              long n, float scale) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride)
    err[p] = mask[p] ? f2bf(bf2f(err[p]) * scale) : (unsigned short)0;
}

void dropout_fwd_launch(const unsigned short* x, unsigned short* y,
                        unsigned char* mask, long n, float ratio,
                        unsigned long long seed, unsigned long long* ctr,
                        hipStream_t s) {
  int blocks = (int)min((n + 255) / 256, (long)4096);
  float scale = 1.f / (1.f - ratio);
  if (ctr)
    hipLaunchKernelGGL(ctr_bump_k, dim3(1), dim3(64), 0, s, ctr);
  hipLaunchKernelGGL(dropout_fwd_k, dim3(blocks), dim3(256), 0, s, x, y,
                     mask, n, ratio, scale, seed, ctr);
  HIP_CHECK_LAST();
}

void dropout_bwd_launch(unsigned short* err, const unsigned char* mask,
                        long n, float ratio, hipStream_t s) {
  int blocks = (int)min((n + 255) / 256, (long)4096);
  hipLaunchKernelGGL(dropout_bwd_k, dim3(blocks), dim3(256), 0, s, err,
                     mask, n, 1.f / (1.f - ratio));
  HIP_CHECK_LAST();
}

// ---------------------------------------------------------------------------
// LDS histogram (the reference's ocl/accumulator.cl
// range/hist kernels) — per-block int32 bins in LDS, one global atomic
// per (block, bin) flush; bf16 and f32 inputs.
// ---------------------------------------------------------------------------
template <typename T>
__global__ void hist_k(const T* __restrict__ x, long n, float lo,
                       float hi, float inv_width, int nbins,
                       unsigned int* __restrict__ out) {
  extern __shared__ unsigned int bins[];
  for (int b = threadIdx.x; b < nbins; b += blockDim.x) bins[b] = 0u;
  __syncthreads();
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long p = i; p < n; p += stride) {
    float v;
    if constexpr (sizeof(T) == 2)
      v = bf2f(((const unsigned short*)x)[p]);
    else
      v = ((const float*)x)[p];
    int b = (int)((v - lo) * inv_width);
    // numpy.histogram semantics: the right edge is inclusive (lands in
    // the last bin); out-of-range values are dropped
    if (v == hi) b = nbins - 1;
    if (v >= lo && b >= 0 && b < nbins) atomicAdd(&bins[b], 1u);
  }
  __syncthreads();
  for (int b = threadIdx.x; b < nbins; b += blockDim.x)
    if (bins[b]) atomicAdd(&out[b], bins[b]);
}

void histogram_launch(const void* x, bool is_bf16, long n, float lo,
                      float hi, int nbins, unsigned int* out,
                      hipStream_t s) {
  if (hi <= lo) hi = lo + 1e-8f;
  const float inv_width = (float)nbins / (hi - lo);
  int blocks = (int)min((n + 255) / 256, (long)2048);
  const size_t lds = (size_t)nbins * 4;
  if (is_bf16)
    hipLaunchKernelGGL(hist_k<unsigned short>, dim3(blocks), dim3(256),
                       lds, s, (const unsigned short*)x, n, lo, hi,
                       inv_width, nbins, out);
  else
    hipLaunchKernelGGL(hist_k<float>, dim3(blocks), dim3(256), lds, s,
                       (const float*)x, n, lo, hi, inv_width, nbins,
                       out);
  HIP_CHECK_LAST();
}

// ---------------------------------------------------------------------------
// depooling over u8 RELATIVE winner codes (pool offsets are no longer
// absolute indices): scatter = unpooling forward, gather = its
// backward; o enumerates the POOL-OUTPUT domain [N,OH,OW,C].
// ---------------------------------------------------------------------------
__global__ void depool_scatter_k(const unsigned short* __restrict__ src,
                                 const unsigned char* __restrict__ offs,
                                 unsigned short* __restrict__ dst,
                                 PoolDesc d, long n, DivM mC, DivM mOW,
                                 DivM mOH, DivM mKX) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long o = i; o < n; o += stride) {
    const int r = offs[o];
    if (r == 255) continue;
    unsigned t = divq((unsigned)o, mC);              // o / C
    unsigned c = (unsigned)o - t * d.C;
    unsigned t2 = divq(t, mOW);
    unsigned ox = t - t2 * d.OW;
    unsigned nb = divq(t2, mOH);
    unsigned oy = t2 - nb * d.OH;
    unsigned ky = divq((unsigned)r, mKX);
    unsigned kx = (unsigned)r - ky * d.KX;
    const long dsti =
        (((long)nb * d.H + oy * d.SY + ky) * d.W + ox * d.SX + kx) *
            d.C + c;
    dst[dsti] = src[o];
  }
}

__global__ void depool_gather_k(const unsigned short* __restrict__ src,
                                const unsigned char* __restrict__ offs,
                                unsigned short* __restrict__ dst,
                                PoolDesc d, long n, DivM mC, DivM mOW,
                                DivM mOH, DivM mKX) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long o = i; o < n; o += stride) {
    const int r = offs[o];
    if (r == 255) {
      dst[o] = 0;
      continue;
    }
    unsigned t = divq((unsigned)o, mC);
    unsigned c = (unsigned)o - t * d.C;
    unsigned t2 = divq(t, mOW);
    unsigned ox = t - t2 * d.OW;
    unsigned nb = divq(t2, mOH);
    unsigned oy = t2 - nb * d.OH;
    unsigned ky = divq((unsigned)r, mKX);
    unsigned kx = (unsigned)r - ky * d.KX;
    const long srci =
        (((long)nb * d.H + oy * d.SY + ky) * d.W + ox * d.SX + kx) *
            d.C + c;
    dst[o] = src[srci];
  }
}

void depool_scatter_launch(const unsigned short* src,
                           const unsigned char* offs,
                           unsigned short* dst, const PoolDesc& d,
                           hipStream_t s) {
  const long n = (long)d.N * d.OH * d.OW * d.C;
  const long big = (long)d.N * d.H * d.W * d.C;
  void zero_f32_launch(float*, long, hipStream_t);
  zero_f32_launch((float*)dst, big / 2, s);    // bf16 numel is even
  int blocks = (int)min((n + 255) / 256, (long)4096);
  hipLaunchKernelGGL(depool_scatter_k, dim3(blocks), dim3(256), 0, s,
                     src, offs, dst, d, n, make_divm((unsigned)d.C),
                     make_divm((unsigned)d.OW), make_divm((unsigned)d.OH),
                     make_divm((unsigned)d.KX));
  HIP_CHECK_LAST();
}

void depool_gather_launch(const unsigned short* src,
                          const unsigned char* offs, unsigned short* dst,
                          const PoolDesc& d, hipStream_t s) {
  const long n = (long)d.N * d.OH * d.OW * d.C;
  int blocks = (int)min((n + 255) / 256, (long)4096);
  hipLaunchKernelGGL(depool_gather_k, dim3(blocks), dim3(256), 0, s,
                     src, offs, dst, d, n, make_divm((unsigned)d.C),
                     make_divm((unsigned)d.OW), make_divm((unsigned)d.OH),
                     make_divm((unsigned)d.KX));
  HIP_CHECK_LAST();
}
