#include "hip/hip_runtime.h"
// Row softmax + loss evaluator kernels (fp32 logits path).
// Replaces reference znicz/ocl/softmax.cl (apply_exp) and evaluator.cl
// (evaluate_softmax / evaluate_mse) — redesigned as wave-level rowwise
// kernels: one wave64 per row, shuffle reductions, device-resident n_err.
#include "common.h"

// in-place softmax on x[M][N] fp32; argmax -> max_idx[M]
__global__ void softmax_k(float* __restrict__ x, int* __restrict__ max_idx,
                          int M, int N) {
  const int row = blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= M) return;
  float* p = x + (long)row * N;
  // wave max + argmax
  float m = -3.4e38f;
  int mi = 0;
  for (int j = lane; j < N; j += WAVE) {
    float v = p[j];
    if (v > m) { m = v; mi = j; }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    float om = __shfl_down(m, off);
    int omi = __shfl_down(mi, off);
    if (om > m || (om == m && omi < mi)) { m = om; mi = omi; }
  }
  m = __shfl(m, 0);
  if (lane == 0 && max_idx) max_idx[row] = mi;
  // exp + sum
  float s = 0.f;
  for (int j = lane; j < N; j += WAVE) {
    float e = __expf(p[j] - m);
    p[j] = e;
    s += e;
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) s += __shfl_down(s, off);
  s = __shfl(s, 0);
  const float inv = 1.f / s;
  for (int j = lane; j < N; j += WAVE) p[j] *= inv;
}

void softmax_launch(float* x, int* max_idx, int M, int N, hipStream_t s) {
  const int waves_per_block = 4;
  dim3 grid((M + waves_per_block - 1) / waves_per_block),
      block(waves_per_block * WAVE);
  hipLaunchKernelGGL(softmax_k, grid, block, 0, s, x, max_idx, M, N);
  HIP_CHECK_LAST();
}

// err[M][N](bf16) = probs - onehot(label) for rows < batch (0 beyond);
// n_err[0] += #(argmax != label); labels < 0 are ignored samples.
__global__ void eval_softmax_k(const float* __restrict__ probs,
                               const int* __restrict__ labels,
                               unsigned short* __restrict__ err,
                               int* __restrict__ n_err, int M, int N,
                               int batch) {
  const int row = blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= M) return;
  unsigned short* e = err + (long)row * N;
  if (row >= batch) {
    for (int j = lane; j < N; j += WAVE) e[j] = 0;
    return;
  }
  const int lbl = labels[row];
  const float* p = probs + (long)row * N;
  float m = -3.4e38f;
  int mi = 0;
  for (int j = lane; j < N; j += WAVE) {
    float v = p[j];
    e[j] = f2bf(lbl >= 0 ? (v - (j == lbl ? 1.f : 0.f)) : 0.f);
    if (v > m) { m = v; mi = j; }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    float om = __shfl_down(m, off);
    int omi = __shfl_down(mi, off);
    if (om > m || (om == m && omi < mi)) { m = om; mi = omi; }
  }
  if (lane == 0 && lbl >= 0 && mi != lbl) atomicAdd(n_err, 1);
}

void eval_softmax_launch(const float* probs, const int* labels,
                         unsigned short* err, int* n_err, int M, int N,
                         int batch, hipStream_t s) {
  const int wpb = 4;
  dim3 grid((M + wpb - 1) / wpb), block(wpb * WAVE);
  hipLaunchKernelGGL(eval_softmax_k, grid, block, 0, s, probs, labels, err,
                     n_err, M, N, batch);
  HIP_CHECK_LAST();
}

// MSE: err = y - t (bf16); mse[row] = (root? sqrt : id)(mean sq);
// metrics[0] += Σ mse ; metrics[1] = max ; metrics[2] = min
__global__ void eval_mse_k(const unsigned short* __restrict__ y,
                           const unsigned short* __restrict__ t,
                           unsigned short* __restrict__ err,
                           float* __restrict__ mse,
                           float* __restrict__ metrics, int M, int N,
                           int batch, int root) {
  const int row = blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= M) return;
  const long off = (long)row * N;
  if (row >= batch) {
    for (int j = lane; j < N; j += WAVE) err[off + j] = 0;
    if (lane == 0) mse[row] = 0.f;
    return;
  }
  float s = 0.f;
  for (int j = lane; j < N; j += WAVE) {
    float d = bf2f(y[off + j]) - bf2f(t[off + j]);
    err[off + j] = f2bf(d);
    s += d * d;
  }
#pragma unroll
  for (int o = 32; o; o >>= 1) s += __shfl_down(s, o);
  if (lane == 0) {
    float v = s / N;
    if (root) v = sqrtf(v);
    mse[row] = v;
    atomicAdd(&metrics[0], v);
    // fp32 max/min via atomicCAS loops are overkill for a per-row call;
    // use integer trick (values are >= 0)
    atomicMax((int*)&metrics[1], __float_as_int(v));
    atomicMin((int*)&metrics[2], __float_as_int(v));
  }
}

void eval_mse_launch(const unsigned short* y, const unsigned short* t,
                     unsigned short* err, float* mse, float* metrics,
                     int M, int N, int batch, int root, hipStream_t s) {
  const int wpb = 4;
  dim3 grid((M + wpb - 1) / wpb), block(wpb * WAVE);
  hipLaunchKernelGGL(eval_mse_k, grid, block, 0, s, y, t, err, mse,
                     metrics, M, N, batch, root);
  HIP_CHECK_LAST();
}
