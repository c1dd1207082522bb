// _mzk — Python bindings for the hand-written CDNA4 kernel set.
// The MI355X-native replacement for the reference's opencl4py/cuda4py
// binding layer: torch tensors in, kernel launches on the CURRENT torch
// stream (so the DP engine's comm-stream overlap works unchanged).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "hip/api.h"

// from gemm.hip
void gemm_nt_dense(const unsigned short*, const unsigned short*, void*,
                   bool, const float*, const unsigned short*, int, int,
                   int, int, hipStream_t,
                   const unsigned short* dact = nullptr);
bool gemm_tn_dense(const unsigned short*, const unsigned short*, float*,
                   int, int, int, float, hipStream_t, float*);
void conv_fwd_gemm(const unsigned short*, const unsigned short*, void*,
                   bool, const float*, const unsigned short*,
                   const ConvDesc&, int, hipStream_t);
void conv_dgrad_gemm(const unsigned short*, const unsigned short*,
                     unsigned short*, const unsigned short*,
                     const ConvDesc&, hipStream_t, int act = 0,
                     const unsigned short* dact = nullptr);
bool conv_wgrad_gemm(const unsigned short*, const unsigned short*, float*,
                     const ConvDesc&, float, hipStream_t, float*);
// from elementwise.hip
void act_fwd_launch(unsigned short*, long, int, hipStream_t);
void act_bwd_launch(unsigned short*, const unsigned short*, long, int,
                    hipStream_t);
void sgd_launch(float*, const float*, float*, unsigned short*, long, float,
                float, float, float, float, hipStream_t);
#include "hip/sgd_multi.h"
void sgd_multi_launch(const void*, const int*, int, hipStream_t);
void col_sum_launch(const unsigned short*, float*, int, int, bool,
                    hipStream_t);
void transpose2d_launch(const unsigned short*, unsigned short*, int, int,
                        hipStream_t);
void permute_kc_launch(const unsigned short*, unsigned short*, int, int,
                       int, hipStream_t);
void gather_off_launch(const unsigned short*, const int*, unsigned short*,
                       long, hipStream_t);
void scatter_off_launch(const unsigned short*, const int*, unsigned short*,
                        long, long, hipStream_t);
void bernoulli_launch(const unsigned short*, unsigned short*, long,
                      unsigned long long, hipStream_t);
void fill_uniform_launch(float*, long, float, float, unsigned long long,
                         hipStream_t);
void row_argmin_launch(const float*, int*, int, int, hipStream_t);
void mean_disp_launch(const unsigned short*, const float*, const float*,
                      unsigned short*, long, long, hipStream_t);
void confusion_launch(const int*, const int*, int*, int, int,
                      hipStream_t);
void dropout_fwd_launch(const unsigned short*, unsigned short*,
                        unsigned char*, long, float, unsigned long long,
                        unsigned long long*, hipStream_t);
void dropout_bwd_launch(unsigned short*, const unsigned char*, long, float,
                        hipStream_t);
void histogram_launch(const void*, bool, long, float, float, int,
                      unsigned int*, hipStream_t);
void set_routing(int, int, int, int);
void set_fc_split(int, int, int, int, int);
// from softmax.hip
void softmax_launch(float*, int*, int, int, hipStream_t);
void eval_softmax_launch(const float*, const int*, unsigned short*, int*,
                         int, int, int, hipStream_t);
void eval_mse_launch(const unsigned short*, const unsigned short*,
                     unsigned short*, float*, float*, int, int, int, int,
                     hipStream_t);
// from pool_lrn.hip
void pool_fwd_launch(const unsigned short*, unsigned short*,
                     unsigned char*, const PoolDesc&, int, hipStream_t);
void depool_scatter_launch(const unsigned short*, const unsigned char*,
                           unsigned short*, const PoolDesc&, hipStream_t);
void depool_gather_launch(const unsigned short*, const unsigned char*,
                          unsigned short*, const PoolDesc&, hipStream_t);
void pool_bwd_launch(const unsigned short*, unsigned short*,
                     const unsigned char*, const PoolDesc&, int,
                     hipStream_t);
void lrn_fwd_launch(const unsigned short*, unsigned short*, float*, long,
                    int, float, float, float, int, hipStream_t);
void lrn_bwd_launch(const unsigned short*, const unsigned short*,
                    const float*, float*, unsigned short*, long, int,
                    float, float, float, int, hipStream_t);

namespace {

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

const unsigned short* ubf(const torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, "expected bf16");
  TORCH_CHECK(t.is_contiguous(), "expected contiguous");
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}

unsigned short* ubf_mut(torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, "expected bf16");
  TORCH_CHECK(t.is_contiguous(), "expected contiguous");
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}

// persistent 1 KiB zero page for padded/out-of-range DMA reads
const unsigned short* zero_page(const torch::Device& dev) {
  static torch::Tensor zp;
  if (!zp.defined() || zp.device() != dev)
    zp = torch::zeros({512}, torch::TensorOptions()
                                 .dtype(torch::kBFloat16)
                                 .device(dev));
  return ubf(zp);
}

ConvDesc make_conv_desc(const std::vector<int64_t>& v) {
  TORCH_CHECK(v.size() == 13, "conv desc needs 13 ints");
  ConvDesc d{(int)v[0], (int)v[1], (int)v[2], (int)v[3], (int)v[4],
             (int)v[5], (int)v[6], (int)v[7], (int)v[8], (int)v[9],
             (int)v[10], (int)v[11], (int)v[12]};
  finish_conv_desc(d);
  return d;
}

void gemm_nt(torch::Tensor a, torch::Tensor b, torch::Tensor out,
             c10::optional<torch::Tensor> bias, int64_t act,
             c10::optional<torch::Tensor> dact) {
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "gemm_nt: K mismatch");
  TORCH_CHECK(out.size(0) == M && out.size(1) == N, "gemm_nt: out shape");
  const float* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32, "bias fp32");
    bp = bias->data_ptr<float>();
  }
  bool f32out = out.scalar_type() == torch::kFloat32;
  void* op = f32out ? (void*)out.data_ptr<float>()
                    : (void*)ubf_mut(out);
  const unsigned short* dp = dact.has_value() ? ubf(*dact) : nullptr;
  if (dp) TORCH_CHECK(dact->numel() == out.numel(), "gemm_nt: dact shape");
  gemm_nt_dense(ubf(a), ubf(b), op, f32out, bp, zero_page(a.device()), M,
                N, K, (int)act, cur_stream(), dp);
}

bool gemm_tn(torch::Tensor a, torch::Tensor b, torch::Tensor out,
             double beta, c10::optional<torch::Tensor> bias_out) {
  // out[M,N] = aᵀ·b with a[K,M], b[K,N]
  const int K = a.size(0), M = a.size(1), N = b.size(1);
  TORCH_CHECK(b.size(0) == K, "gemm_tn: K mismatch");
  TORCH_CHECK(out.scalar_type() == torch::kFloat32, "gemm_tn: fp32 out");
  TORCH_CHECK(out.numel() == (int64_t)M * N, "gemm_tn: out shape");
  float* bp = bias_out.has_value() ? bias_out->data_ptr<float>()
                                   : nullptr;
  // true: the kernel also produced bias_out = column sums of a (the
  // bias gradient); false: caller must col_sum separately
  return gemm_tn_dense(ubf(a), ubf(b), out.data_ptr<float>(), M, N, K,
                       (float)beta, cur_stream(), bp);
}

void conv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor out,
              c10::optional<torch::Tensor> bias,
              std::vector<int64_t> desc, int64_t act) {
  ConvDesc d = make_conv_desc(desc);
  const float* bp = nullptr;
  if (bias.has_value()) bp = bias->data_ptr<float>();
  bool f32out = out.scalar_type() == torch::kFloat32;
  void* op = f32out ? (void*)out.data_ptr<float>()
                    : (void*)ubf_mut(out);
  conv_fwd_gemm(ubf(x), ubf(w), op, f32out, bp, zero_page(x.device()), d,
                (int)act, cur_stream());
}

void conv_dgrad(torch::Tensor eo, torch::Tensor wdg, torch::Tensor ei,
                std::vector<int64_t> desc, int64_t act,
                c10::optional<torch::Tensor> dact) {
  ConvDesc d = make_conv_desc(desc);
  const unsigned short* dp = dact.has_value() ? ubf(*dact) : nullptr;
  if (dp) TORCH_CHECK(dact->numel() == ei.numel(),
                      "conv_dgrad: dact shape");
  conv_dgrad_gemm(ubf(eo), ubf(wdg), ubf_mut(ei), zero_page(eo.device()),
                  d, cur_stream(), (int)act, dp);
}

bool conv_wgrad(torch::Tensor eo, torch::Tensor x, torch::Tensor gw,
                std::vector<int64_t> desc, double beta,
                c10::optional<torch::Tensor> bias_out) {
  ConvDesc d = make_conv_desc(desc);
  float* bp = bias_out.has_value() ? bias_out->data_ptr<float>()
                                   : nullptr;
  return conv_wgrad_gemm(ubf(eo), ubf(x), gw.data_ptr<float>(), d,
                         (float)beta, cur_stream(), bp);
}

void act_fwd(torch::Tensor x, int64_t act) {
  act_fwd_launch(ubf_mut(x), x.numel(), (int)act, cur_stream());
}

void act_bwd(torch::Tensor err, torch::Tensor y, int64_t act) {
  TORCH_CHECK(err.numel() == y.numel(), "act_bwd size");
  act_bwd_launch(ubf_mut(err), ubf(y), err.numel(), (int)act,
                 cur_stream());
}

void sgd_step(torch::Tensor w, torch::Tensor g, torch::Tensor vel,
              c10::optional<torch::Tensor> wb,
              c10::optional<torch::Tensor> wbt, double lr, double decay,
              double l1, double moment, double scale) {
  unsigned short* wbp = nullptr;
  if (wb.has_value()) wbp = ubf_mut(*wb);
  sgd_launch(w.data_ptr<float>(), g.data_ptr<float>(),
             vel.data_ptr<float>(), wbp, w.numel(), (float)lr,
             (float)decay, (float)l1, (float)moment, (float)scale,
             cur_stream());
  if (wbt.has_value()) {
    TORCH_CHECK(wb.has_value() && w.dim() == 2, "wbt needs 2-D wb");
    transpose2d_launch(ubf(*wb), ubf_mut(*wbt), w.size(0), w.size(1),
                       cur_stream());
  }
}

// -- multi-tensor SGD (one launch per gradient bucket) ----------------------
// Build the descriptor blob + block->tensor table on the host (CPU
// tensors; the caller uploads them once and reuses across steps —
// hipGraph replays then read only device memory).
std::vector<torch::Tensor> sgd_multi_prepare(
    std::vector<torch::Tensor> w, std::vector<torch::Tensor> g,
    std::vector<torch::Tensor> vel,
    std::vector<c10::optional<torch::Tensor>> wb,
    std::vector<double> lr, std::vector<double> decay,
    std::vector<double> l1, std::vector<double> moment,
    std::vector<double> scale) {
  const size_t nt = w.size();
  TORCH_CHECK(g.size() == nt && vel.size() == nt && wb.size() == nt &&
              lr.size() == nt, "sgd_multi_prepare: length mismatch");
  std::vector<SgdDesc> descs(nt);
  std::vector<int> b2t;
  long long blk = 0;
  for (size_t i = 0; i < nt; ++i) {
    SgdDesc& d = descs[i];
    d.w = w[i].data_ptr<float>();
    d.g = g[i].data_ptr<float>();
    d.vel = vel[i].data_ptr<float>();
    d.wb = wb[i].has_value() ? ubf_mut(*wb[i]) : nullptr;
    d.n = w[i].numel();
    d.blk0 = blk;
    d.lr = (float)lr[i];
    d.decay = (float)decay[i];
    d.l1 = (float)l1[i];
    d.moment = (float)moment[i];
    d.scale = (float)scale[i];
    const long long nb = (d.n + SGD_MULTI_SPAN - 1) / SGD_MULTI_SPAN;
    for (long long b = 0; b < nb; ++b) b2t.push_back((int)i);
    blk += nb;
  }
  auto blob = torch::empty({(long)(nt * sizeof(SgdDesc))},
                           torch::dtype(torch::kUInt8));
  memcpy(blob.data_ptr<uint8_t>(), descs.data(), nt * sizeof(SgdDesc));
  auto tbl = torch::empty({(long)b2t.size()},
                          torch::dtype(torch::kInt32));
  memcpy(tbl.data_ptr<int>(), b2t.data(), b2t.size() * sizeof(int));
  return {blob, tbl};
}

void sgd_multi(torch::Tensor desc_blob_dev, torch::Tensor b2t_dev) {
  TORCH_CHECK(desc_blob_dev.is_cuda() && b2t_dev.is_cuda(),
              "sgd_multi: device tensors required");
  sgd_multi_launch(desc_blob_dev.data_ptr<uint8_t>(),
                   b2t_dev.data_ptr<int>(), (int)b2t_dev.numel(),
                   cur_stream());
}

void permute_kc(torch::Tensor src, torch::Tensor dst, int64_t K,
                int64_t P, int64_t C) {
  TORCH_CHECK(src.numel() == K * P * C && dst.numel() == src.numel(),
              "permute_kc shape");
  permute_kc_launch(ubf(src), ubf_mut(dst), (int)K, (int)P, (int)C,
                    cur_stream());
}

void col_sum(torch::Tensor a, torch::Tensor out, bool accumulate) {
  col_sum_launch(ubf(a), out.data_ptr<float>(), a.size(0), a.size(1),
                 accumulate, cur_stream());
}

void transpose2d(torch::Tensor a, torch::Tensor out) {
  transpose2d_launch(ubf(a), ubf_mut(out), a.size(0), a.size(1),
                     cur_stream());
}

void softmax_fwd(torch::Tensor x, torch::Tensor max_idx) {
  softmax_launch(x.data_ptr<float>(), max_idx.data_ptr<int>(), x.size(0),
                 x.size(1), cur_stream());
}

void eval_softmax(torch::Tensor probs, torch::Tensor labels,
                  torch::Tensor err, torch::Tensor n_err, int64_t batch) {
  eval_softmax_launch(probs.data_ptr<float>(), labels.data_ptr<int>(),
                      ubf_mut(err), n_err.data_ptr<int>(), probs.size(0),
                      probs.size(1), (int)batch, cur_stream());
}

void eval_mse(torch::Tensor y, torch::Tensor t, torch::Tensor err,
              torch::Tensor mse, torch::Tensor metrics, int64_t batch,
              int64_t root) {
  eval_mse_launch(ubf(y), ubf(t), ubf_mut(err), mse.data_ptr<float>(),
                  metrics.data_ptr<float>(), y.size(0), y.size(1),
                  (int)batch, (int)root, cur_stream());
}

void pool_fwd(torch::Tensor x, torch::Tensor y, torch::Tensor offs,
              std::vector<int64_t> v, int64_t mode) {
  TORCH_CHECK(v.size() == 11, "pool desc needs 11 ints");
  PoolDesc d{(int)v[0], (int)v[1], (int)v[2], (int)v[3], (int)v[4],
             (int)v[5], (int)v[6], (int)v[7], (int)v[8], (int)v[9],
             (unsigned long long)v[10]};
  TORCH_CHECK(offs.dtype() == torch::kUInt8,
              "pool offsets are u8 relative window codes");
  TORCH_CHECK(d.KY * d.KX <= 254, "pool window too large for u8 codes");
  pool_fwd_launch(ubf(x), ubf_mut(y), offs.data_ptr<uint8_t>(), d,
                  (int)mode, cur_stream());
}

void pool_bwd(torch::Tensor ey, torch::Tensor ex, torch::Tensor offs,
              std::vector<int64_t> v, int64_t mode) {
  TORCH_CHECK(v.size() == 11, "pool desc needs 11 ints");
  PoolDesc d{(int)v[0], (int)v[1], (int)v[2], (int)v[3], (int)v[4],
             (int)v[5], (int)v[6], (int)v[7], (int)v[8], (int)v[9],
             (unsigned long long)v[10]};
  TORCH_CHECK(offs.dtype() == torch::kUInt8,
              "pool offsets are u8 relative window codes");
  pool_bwd_launch(ubf(ey), ubf_mut(ex), offs.data_ptr<uint8_t>(), d,
                  (int)mode, cur_stream());
}

static PoolDesc make_pool_desc(const std::vector<int64_t>& v) {
  TORCH_CHECK(v.size() == 11, "pool desc needs 11 ints");
  return PoolDesc{(int)v[0], (int)v[1], (int)v[2], (int)v[3], (int)v[4],
                  (int)v[5], (int)v[6], (int)v[7], (int)v[8], (int)v[9],
                  (unsigned long long)v[10]};
}

void depool_scatter(torch::Tensor src, torch::Tensor offs,
                    torch::Tensor dst, std::vector<int64_t> v) {
  PoolDesc d = make_pool_desc(v);
  TORCH_CHECK(offs.dtype() == torch::kUInt8, "u8 offsets");
  depool_scatter_launch(ubf(src), offs.data_ptr<uint8_t>(), ubf_mut(dst),
                        d, cur_stream());
}

void depool_gather(torch::Tensor src, torch::Tensor offs,
                   torch::Tensor dst, std::vector<int64_t> v) {
  PoolDesc d = make_pool_desc(v);
  TORCH_CHECK(offs.dtype() == torch::kUInt8, "u8 offsets");
  depool_gather_launch(ubf(src), offs.data_ptr<uint8_t>(), ubf_mut(dst),
                       d, cur_stream());
}

void lrn_fwd(torch::Tensor x, torch::Tensor y,
             c10::optional<torch::Tensor> denom, double alpha,
             double beta, double k, int64_t win) {
  const int C = x.size(-1);
  // denom is optional: the backward recomputes it from x, so the hot
  // path skips the fp32 round-trip entirely
  float* dp = denom.has_value() ? denom->data_ptr<float>() : nullptr;
  lrn_fwd_launch(ubf(x), ubf_mut(y), dp, x.numel() / C, C, (float)alpha,
                 (float)beta, (float)k, (int)win, cur_stream());
}

void lrn_bwd(torch::Tensor ey, torch::Tensor x,
             c10::optional<torch::Tensor> y,
             c10::optional<torch::Tensor> denom, torch::Tensor ex,
             double alpha, double beta, double k, int64_t win) {
  const int C = x.size(-1);
  (void)y;
  const float* dp = denom.has_value() ? denom->data_ptr<float>()
                                      : nullptr;
  lrn_bwd_launch(ubf(ey), ubf(x), dp, nullptr, ubf_mut(ex),
                 x.numel() / C, C, (float)alpha, (float)beta, (float)k,
                 (int)win, cur_stream());
}

void dropout_fwd(torch::Tensor x, torch::Tensor y, torch::Tensor mask,
                 double ratio, int64_t seed, int64_t counter,
                 c10::optional<torch::Tensor> ctr_buf) {
  // with ctr_buf the step counter lives on the device (int64[1]) and the
  // kernel sequence is hipGraph-replayable; otherwise the host counter
  // is folded into the seed
  unsigned long long s = (unsigned long long)seed * 0x100000001B3ull;
  unsigned long long* ctr = nullptr;
  if (ctr_buf.has_value())
    ctr = (unsigned long long*)ctr_buf->data_ptr<int64_t>();
  else
    s += (unsigned long long)counter;
  dropout_fwd_launch(ubf(x), ubf_mut(y), mask.data_ptr<unsigned char>(),
                     x.numel(), (float)ratio, s, ctr, cur_stream());
}

void dropout_bwd(torch::Tensor err, torch::Tensor mask, double ratio) {
  dropout_bwd_launch(ubf_mut(err), mask.data_ptr<unsigned char>(),
                     err.numel(), (float)ratio, cur_stream());
}

void gather_by_offset(torch::Tensor src, torch::Tensor offs,
                      torch::Tensor dst) {
  gather_off_launch(ubf(src), offs.data_ptr<int>(), ubf_mut(dst),
                    dst.numel(), cur_stream());
}

void scatter_by_offset(torch::Tensor src, torch::Tensor offs,
                       torch::Tensor dst) {
  scatter_off_launch(ubf(src), offs.data_ptr<int>(), ubf_mut(dst),
                     src.numel(), dst.numel(), cur_stream());
}

void bernoulli(torch::Tensor x, torch::Tensor y, int64_t seed) {
  bernoulli_launch(ubf(x), ubf_mut(y), x.numel(),
                   (unsigned long long)seed, cur_stream());
}

void fill_uniform(torch::Tensor y, double lo, double hi, int64_t seed) {
  fill_uniform_launch(y.data_ptr<float>(), y.numel(), (float)lo,
                      (float)hi, (unsigned long long)seed, cur_stream());
}

void row_argmin(torch::Tensor d, torch::Tensor winners) {
  row_argmin_launch(d.data_ptr<float>(), winners.data_ptr<int>(),
                    d.size(0), d.size(1), cur_stream());
}

void mean_disp(torch::Tensor x, torch::Tensor mean, torch::Tensor rdisp,
               torch::Tensor y) {
  mean_disp_launch(ubf(x), mean.data_ptr<float>(),
                   rdisp.data_ptr<float>(), ubf_mut(y), x.numel(),
                   mean.numel(), cur_stream());
}

void set_routing_py(int64_t nt_split, int64_t tn_split,
                    int64_t tn2_split, int64_t ntn_max_n) {
  set_routing((int)nt_split, (int)tn_split, (int)tn2_split,
              (int)ntn_max_n);
}

void set_fc_split_py(int64_t is_tn, int64_t M, int64_t N, int64_t K,
                     int64_t splits) {
  set_fc_split((int)is_tn, (int)M, (int)N, (int)K, (int)splits);
}

void histogram(torch::Tensor x, torch::Tensor counts, double lo,
               double hi) {
  TORCH_CHECK(x.is_cuda() && counts.is_cuda(), "histogram: cuda only");
  TORCH_CHECK(counts.dtype() == torch::kInt32, "counts must be int32");
  TORCH_CHECK(x.is_contiguous() && counts.is_contiguous());
  const bool bf = x.dtype() == torch::kBFloat16;
  TORCH_CHECK(bf || x.dtype() == torch::kFloat32,
              "histogram: bf16 or f32 input");
  histogram_launch(x.data_ptr(), bf, x.numel(), (float)lo, (float)hi,
                   (int)counts.numel(),
                   (unsigned int*)counts.data_ptr<int>(), cur_stream());
}

void confusion_accum(torch::Tensor max_idx, torch::Tensor labels,
                     torch::Tensor cm, int64_t batch) {
  confusion_launch(max_idx.data_ptr<int>(), labels.data_ptr<int>(),
                   cm.data_ptr<int>(), (int)batch, cm.size(0),
                   cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("confusion_accum", &confusion_accum);
  m.def("gather_by_offset", &gather_by_offset);
  m.def("scatter_by_offset", &scatter_by_offset);
  m.def("bernoulli", &bernoulli);
  m.def("fill_uniform", &fill_uniform);
  m.def("row_argmin", &row_argmin);
  m.def("mean_disp", &mean_disp);
  m.doc() = "veles.znicz_amd hand-written CDNA4 (gfx950) kernels";
  m.def("gemm_nt", &gemm_nt, "C=act(A·Bᵀ+bias), bf16 MFMA",
        py::arg("a"), py::arg("b"), py::arg("out"),
        py::arg("bias") = c10::nullopt, py::arg("act") = 0,
        py::arg("dact") = c10::nullopt);
  m.def("gemm_tn", &gemm_tn, "C=AᵀB (K-major wgrad core)");
  m.def("conv_fwd", &conv_fwd, "implicit-im2col conv forward");
  m.def("conv_dgrad", &conv_dgrad, "conv backward-data (gather)",
        py::arg("eo"), py::arg("wdg"), py::arg("ei"), py::arg("desc"),
        py::arg("act") = 0, py::arg("dact") = c10::nullopt);
  m.def("conv_wgrad", &conv_wgrad, "conv backward-weights (K-major)");
  m.def("act_fwd", &act_fwd);
  m.def("act_bwd", &act_bwd);
  m.def("sgd_step", &sgd_step);
  m.def("sgd_multi_prepare", &sgd_multi_prepare);
  m.def("sgd_multi", &sgd_multi);
  m.def("permute_kc", &permute_kc);
  m.def("col_sum", &col_sum);
  m.def("transpose2d", &transpose2d);
  m.def("softmax_fwd", &softmax_fwd);
  m.def("eval_softmax", &eval_softmax);
  m.def("eval_mse", &eval_mse);
  m.def("pool_fwd", &pool_fwd);
  m.def("pool_bwd", &pool_bwd);
  m.def("depool_scatter", &depool_scatter);
  m.def("depool_gather", &depool_gather);
  m.def("lrn_fwd", &lrn_fwd);
  m.def("lrn_bwd", &lrn_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("histogram", &histogram, "LDS histogram (counts int32)");
  m.def("set_routing", &set_routing_py,
        "override GEMM routing constants (gfx950_routing.json)");
  m.def("set_fc_split", &set_fc_split_py,
        "per-shape dense-GEMM split override: (is_tn, M, N, K, splits);"
        " splits<=0 removes the row");
}
