// Standalone probe: column-sum variants on MI355X.
//   hipcc --offload-arch=gfx950 -O3 tools/probe_colsum.hip -o /tmp/probe
// Finds why the two-pass col_sum sits at ~250 us where torch's reduce
// does 52 us (M=774400, N=96).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>

#define CHECK(x) do { hipError_t e = (x); if (e) { \
  printf("ERR %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } \
} while (0)

typedef __attribute__((ext_vector_type(8))) short short8;

__device__ __forceinline__ float bf2f(unsigned short h) {
  union { unsigned int u; float f; } v;
  v.u = ((unsigned int)h) << 16;
  return v.f;
}

// V1: current design — pow2 octet mapping, one 16B load per iteration
__global__ void v1(const unsigned short* __restrict__ a, float* ws,
                   int M, int N, int rps, int lgo) {
  const int O2 = 1 << lgo;
  const int oct = threadIdx.x & (O2 - 1);
  const int rsub = threadIdx.x >> lgo;
  const int rstep = blockDim.x >> lgo;
  const int cb = oct * 8;
  float s[8] = {};
  const long m0 = (long)blockIdx.x * rps;
  const long m1 = min((long)M, m0 + rps);
  if (cb < N) {
    const unsigned short* p = a + (m0 + rsub) * (long)N + cb;
    for (long m = m0 + rsub; m < m1; m += rstep, p += (long)N * rstep) {
      unsigned short v[8];
      *(short8*)v = *(const short8*)p;
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += bf2f(v[j]);
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
    float* dst = ws + (long)blockIdx.x * N + cb;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (cb + j < N) dst[j] = s[j];
  }
}

// V2: flat grid-stride over ELEMENTS, per-thread partial into LDS by
// (col % N) — no: torch-style: each thread sums a strided set of FULL
// rows worth of contiguous elements, block-local LDS histogram by col.
// Simpler: each thread processes contiguous 8-elem chunks grid-stride;
// chunk's column block = (chunk*8 % N) — requires N % 8 == 0 and
// N ≤ 2048.  Partials in LDS [N] per block, atomic-free ws write.
extern __shared__ float lds_cols[];
__global__ void v2(const unsigned short* __restrict__ a, float* ws,
                   int M, int N, long total8) {
  for (int c = threadIdx.x; c < N; c += blockDim.x) lds_cols[c] = 0.f;
  __syncthreads();
  const int noct = N >> 3;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float s[8] = {};
  int my_oct = -2, cur_oct;
  for (long q = i; q < total8; q += stride) {
    cur_oct = (int)(q % noct);          // column octet of this chunk
    if (cur_oct != my_oct) {
      if (my_oct >= 0)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          atomicAdd(&lds_cols[my_oct * 8 + j], s[j]);
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] = 0.f;
      my_oct = cur_oct;
    }
    unsigned short v[8];
    *(short8*)v = *(const short8*)(a + q * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) s[j] += bf2f(v[j]);
  }
  if (my_oct >= 0)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      atomicAdd(&lds_cols[my_oct * 8 + j], s[j]);
  __syncthreads();
  for (int c = threadIdx.x; c < N; c += blockDim.x)
    ws[(long)blockIdx.x * N + c] = lds_cols[c];
}

// V3: like V1 but 4 independent row-streams per thread (ILP) — each
// thread strides 4*rstep, keeping 4 loads in flight.
__global__ void v3(const unsigned short* __restrict__ a, float* ws,
                   int M, int N, int rps, int lgo) {
  const int O2 = 1 << lgo;
  const int oct = threadIdx.x & (O2 - 1);
  const int rsub = threadIdx.x >> lgo;
  const int rstep = blockDim.x >> lgo;
  const int cb = oct * 8;
  float s[8] = {};
  const long m0 = (long)blockIdx.x * rps;
  const long m1 = min((long)M, m0 + rps);
  if (cb < N) {
    const long step = (long)rstep;
    long m = m0 + rsub;
    const unsigned short* p = a + m * (long)N + cb;
    const long stride_b = (long)N * step;
    // 4-deep manual pipeline
    for (; m + 3 * step < m1; m += 4 * step, p += 4 * stride_b) {
      unsigned short v0[8], v1_[8], v2_[8], v3_[8];
      *(short8*)v0 = *(const short8*)p;
      *(short8*)v1_ = *(const short8*)(p + stride_b);
      *(short8*)v2_ = *(const short8*)(p + 2 * stride_b);
      *(short8*)v3_ = *(const short8*)(p + 3 * stride_b);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s[j] += bf2f(v0[j]) + bf2f(v1_[j]) + bf2f(v2_[j]) +
                bf2f(v3_[j]);
    }
    for (; m < m1; m += step, p += stride_b) {
      unsigned short v[8];
      *(short8*)v = *(const short8*)p;
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += bf2f(v[j]);
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
    float* dst = ws + (long)blockIdx.x * N + cb;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (cb + j < N) dst[j] = s[j];
  }
}

__global__ void fin(const float* ws, float* out, int slabs, int N) {
  int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= N) return;
  float s = 0.f;
  for (int r = 0; r < slabs; ++r) s += ws[(long)r * N + col];
  out[col] = s;
}

// pure bandwidth reference: read the same data contiguously
__global__ void bw(const unsigned short* __restrict__ a, float* out,
                   long total8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float s = 0.f;
  for (long q = i; q < total8; q += stride) {
    unsigned short v[8];
    *(short8*)v = *(const short8*)(a + q * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) s += bf2f(v[j]);
  }
  if (s == 1e30f) out[0] = s;   // never true; keep the work
}

static float time_kernel(void (*launch)(int), int arg, int reps) {
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  launch(arg);
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int r = 0; r < reps; ++r) launch(arg);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  return ms / reps * 1000.f;
}

const unsigned short* g_a;
float *g_ws, *g_out;
int g_M, g_N;

static int lgo_of(int N) {
  int noct = (N + 7) >> 3, lg = 0;
  while ((1 << lg) < noct && lg < 8) ++lg;
  return lg;
}

int main() {
  struct { int M, N; } shapes[] = {{774400, 96}, {186624, 256},
                                   {43264, 384}};
  for (auto& sh : shapes) {
    int M = sh.M, N = sh.N;
    g_M = M; g_N = N;
    unsigned short* a;
    CHECK(hipMalloc(&a, (size_t)M * N * 2));
    CHECK(hipMemset(a, 0x3f, (size_t)M * N * 2));
    CHECK(hipMalloc(&g_ws, (size_t)8192 * N * 4));
    CHECK(hipMalloc(&g_out, N * 4));
    g_a = a;
    double gb = (double)M * N * 2 / 1e9;
    printf("-- M=%d N=%d (%.0f MB)\n", M, N, gb * 1000);

    for (int slabs : {512, 1024, 2048, 4096, 8192}) {
      int rps = (M + slabs - 1) / slabs;
      int lgo = lgo_of(N);
      auto launch = +[](int s) {
        int rps_ = (g_M + s - 1) / s;
        hipLaunchKernelGGL(v1, dim3(s), dim3(256), 0, 0, g_a, g_ws,
                           g_M, g_N, rps_, lgo_of(g_N));
        hipLaunchKernelGGL(fin, dim3((g_N + 255) / 256), dim3(256), 0,
                           0, g_ws, g_out, s, g_N);
      };
      float us = time_kernel(launch, slabs, 20);
      printf("  v1 slabs=%5d rps=%5d: %7.1f us (%5.0f GB/s)\n", slabs,
             rps, us, gb / us * 1e6);
    }
    for (int slabs : {1024, 4096}) {
      auto launch = +[](int s) {
        int rps_ = (g_M + s - 1) / s;
        hipLaunchKernelGGL(v3, dim3(s), dim3(256), 0, 0, g_a, g_ws,
                           g_M, g_N, rps_, lgo_of(g_N));
        hipLaunchKernelGGL(fin, dim3((g_N + 255) / 256), dim3(256), 0,
                           0, g_ws, g_out, s, g_N);
      };
      float us = time_kernel(launch, slabs, 20);
      printf("  v3 slabs=%5d (ILP4):   %7.1f us (%5.0f GB/s)\n", slabs,
             us, gb / us * 1e6);
    }
    if (N % 8 == 0) {
      for (int blocks : {1024, 2048}) {
        auto launch = +[](int b) {
          long total8 = (long)g_M * g_N / 8;
          hipLaunchKernelGGL(v2, dim3(b), dim3(256), g_N * 4, 0, g_a,
                             g_ws, g_M, g_N, total8);
          hipLaunchKernelGGL(fin, dim3((g_N + 255) / 256), dim3(256),
                             0, 0, g_ws, g_out, b, g_N);
        };
        float us = time_kernel(launch, blocks, 20);
        printf("  v2 blocks=%4d (flat):  %7.1f us (%5.0f GB/s)\n",
               blocks, us, gb / us * 1e6);
      }
    }
    {
      auto launch = +[](int b) {
        long total8 = (long)g_M * g_N / 8;
        hipLaunchKernelGGL(bw, dim3(b), dim3(256), 0, 0, g_a, g_out,
                           total8);
      };
      float us = time_kernel(launch, 2048, 20);
      printf("  bw (contig read only):  %7.1f us (%5.0f GB/s)\n", us,
             gb / us * 1e6);
    }
    hipFree(a);
    hipFree(g_ws);
    hipFree(g_out);
  }
  return 0;
}
