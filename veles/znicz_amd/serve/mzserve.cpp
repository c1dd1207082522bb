// mzserve — standalone MI355X inference engine for exported Veles
// forward chains (the native analog of the reference ecosystem's
// libVeles/libZnicz C++ engines, which loaded snapshots for inference
// outside Python — SURVEY §2b).
//
// Input: a directory written by veles.serving.export_flat() holding
//   manifest.txt   line-oriented op list + tensor table
//   weights.bin    raw blob; every tensor already in the device layout
//                  the kernels take (FC [N,K] bf16, conv [K,KH*KW*C]
//                  bf16, fp32 biases), 16-B aligned offsets
// Links the SAME kernel translation units as the _mzk extension
// (gemm.hip / gemm8.hip / pool_lrn.hip / elementwise.hip /
// softmax.hip) — no torch, no Python: hip runtime only.  The whole
// forward pass is captured into one hipGraph after a warmup batch and
// replayed per batch (launch-bound regime), mirroring the Python
// InferenceEngine.
//
// Build (done by __graft_entry__.build() / serve/build.sh):
//   hipcc -O3 -std=c++17 --offload-arch=gfx950 mzserve.cpp \
//     ../ops/hip/{gemm,gemm8,elementwise,softmax,pool_lrn}.hip -o mzserve
//
// Run:
//   ./mzserve <export_dir> [--input in.f32.bin] [--out out.f32.bin]
//             [--batches K] [--warmup W] [--no-graph] [--parse-only]
//             [--h2d]   (end-to-end: per-batch convert+upload timed)
// prints one JSON line: {"img_s":..., "ms_per_batch":..., ...}

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <chrono>
#include <fstream>
#include <map>
#include <random>
#include <sstream>
#include <string>
#include <vector>

#include "../ops/hip/api.h"

// raw-pointer launchers from the kernel TUs (the same prototypes
// bindings.cpp declares; see veles/znicz_amd/ops/bindings.cpp)
void gemm_nt_dense(const unsigned short*, const unsigned short*, void*,
                   bool, const float*, const unsigned short*, int, int,
                   int, int, hipStream_t,
                   const unsigned short* dact = nullptr);
void conv_fwd_gemm(const unsigned short*, const unsigned short*, void*,
                   bool, const float*, const unsigned short*,
                   const ConvDesc&, int, hipStream_t);
void pool_fwd_launch(const unsigned short*, unsigned short*,
                     unsigned char*, const PoolDesc&, int, hipStream_t);
void lrn_fwd_launch(const unsigned short*, unsigned short*, float*, long,
                    int, float, float, float, int, hipStream_t);
void softmax_launch(float*, int*, int, int, hipStream_t);
void act_fwd_launch(unsigned short*, long, int, hipStream_t);

#define HIP_OK(call)                                                     \
  do {                                                                   \
    hipError_t e_ = (call);                                              \
    if (e_ != hipSuccess) {                                              \
      std::fprintf(stderr, "mzserve: %s failed: %s (%s:%d)\n", #call,    \
                   hipGetErrorString(e_), __FILE__, __LINE__);           \
      std::exit(1);                                                      \
    }                                                                    \
  } while (0)

namespace {

// fp32 -> bf16 round-to-nearest-even (matches torch's host convert)
inline unsigned short f2bf(float f) {
  unsigned u;
  std::memcpy(&u, &f, 4);
  if ((u & 0x7F800000u) == 0x7F800000u)      // inf/nan: truncate
    return (unsigned short)(u >> 16);
  u += 0x7FFFu + ((u >> 16) & 1);
  return (unsigned short)(u >> 16);
}

inline float bf2f(unsigned short b) {
  unsigned u = (unsigned)b << 16;
  float f;
  std::memcpy(&f, &u, 4);
  return f;
}

struct TensorRef {
  std::string dtype;
  long numel = 0;
  long offset = 0;
};

struct Op {
  std::string kind;          // fc | softmax | conv | pool | lrn | act
  std::string w, bias;       // tensor names ("-" = none)
  int act = 0;
  int M = 0, N = 0, K = 0;   // fc
  bool out_f32 = false;
  int rows = 0, cols = 0;    // softmax
  std::vector<int> desc;     // conv (13) / pool (11)
  int mode = 0;              // pool
  int C = 0;                 // lrn
  long lrn_rows = 0;
  float alpha = 0, beta = 0, k = 0;
  int win = 0;
  long out_numel = 0;
};

struct Model {
  int batch = 0;
  std::vector<long> in_shape;
  std::map<std::string, TensorRef> tensors;
  std::vector<Op> ops;
};

Model parse_manifest(const std::string& dir) {
  std::ifstream f(dir + "/manifest.txt");
  if (!f) {
    std::fprintf(stderr, "mzserve: cannot open %s/manifest.txt\n",
                 dir.c_str());
    std::exit(1);
  }
  Model m;
  std::string line;
  std::getline(f, line);
  if (line.rfind("mzserve 1", 0) != 0) {
    std::fprintf(stderr, "mzserve: bad manifest header \"%s\"\n",
                 line.c_str());
    std::exit(1);
  }
  while (std::getline(f, line)) {
    std::istringstream ss(line);
    std::string tok;
    ss >> tok;
    if (tok == "end") break;
    if (tok == "batch") {
      ss >> m.batch;
    } else if (tok == "in_shape") {
      long v;
      while (ss >> v) m.in_shape.push_back(v);
    } else if (tok == "tensor") {
      std::string name;
      TensorRef t;
      ss >> name >> t.dtype >> t.numel >> t.offset;
      m.tensors[name] = t;
    } else if (tok == "op") {
      Op op;
      ss >> op.kind;
      if (op.kind == "fc") {
        std::string od;
        ss >> op.w >> op.bias >> op.act >> op.M >> op.N >> op.K >> od;
        op.out_f32 = (od == "f32");
        op.out_numel = (long)op.M * op.N;
      } else if (op.kind == "softmax") {
        ss >> op.rows >> op.cols;
      } else if (op.kind == "conv") {
        ss >> op.w >> op.bias >> op.act;
        op.desc.resize(13);
        for (int& v : op.desc) ss >> v;
        ss >> op.out_numel;
      } else if (op.kind == "pool") {
        ss >> op.mode;
        op.desc.resize(11);
        for (int& v : op.desc) ss >> v;
        ss >> op.out_numel;
      } else if (op.kind == "lrn") {
        ss >> op.C >> op.lrn_rows >> op.alpha >> op.beta >> op.k >>
            op.win >> op.out_numel;
      } else if (op.kind == "act") {
        ss >> op.act;
      } else {
        std::fprintf(stderr, "mzserve: unknown op %s\n",
                     op.kind.c_str());
        std::exit(1);
      }
      m.ops.push_back(op);
    }
  }
  if (!m.batch || m.in_shape.empty() || m.ops.empty()) {
    std::fprintf(stderr, "mzserve: incomplete manifest\n");
    std::exit(1);
  }
  // the first op must produce into its own buffer: softmax/act run
  // in place on the PREVIOUS op's output and would otherwise mutate
  // the resident input across replays
  if (m.ops[0].kind == "act" || m.ops[0].kind == "softmax") {
    std::fprintf(stderr, "mzserve: manifest starts with in-place op "
                         "%s\n", m.ops[0].kind.c_str());
    std::exit(1);
  }
  return m;
}

struct Engine {
  Model m;
  hipStream_t stream = nullptr;
  char* weights = nullptr;           // one device blob; tensors at offs
  unsigned short* zero_page = nullptr;
  unsigned short* input = nullptr;   // bf16 [batch, ...]
  unsigned char* offs = nullptr;     // pool scratch (max over ops)
  int* max_idx = nullptr;            // softmax scratch
  std::vector<void*> outs;           // per-op output buffers
  void* final_buf = nullptr;
  bool final_f32 = false;
  long final_numel = 0;
  long in_numel = 0;

  const void* tptr(const std::string& name) const {
    if (name == "-") return nullptr;
    auto it = m.tensors.find(name);
    if (it == m.tensors.end()) {
      std::fprintf(stderr, "mzserve: unknown tensor %s\n", name.c_str());
      std::exit(1);
    }
    return weights + it->second.offset;
  }

  void setup(const std::string& dir) {
    // weights load + bounds check first: no hip calls yet, so a
    // malformed export fails cleanly even without a GPU
    std::ifstream wf(dir + "/weights.bin", std::ios::binary);
    std::vector<char> blob((std::istreambuf_iterator<char>(wf)),
                           std::istreambuf_iterator<char>());
    for (const auto& kv : m.tensors) {
      const TensorRef& t = kv.second;
      long bytes = t.numel * (t.dtype == "f32" ? 4 : 2);
      if (t.offset < 0 || t.offset + bytes > (long)blob.size()) {
        std::fprintf(stderr,
                     "mzserve: tensor %s [%ld B at %ld] exceeds "
                     "weights.bin (%zu B)\n", kv.first.c_str(), bytes,
                     t.offset, blob.size());
        std::exit(1);
      }
    }
    HIP_OK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    HIP_OK(hipMalloc(&weights, blob.size() ? blob.size() : 16));
    HIP_OK(hipMemcpy(weights, blob.data(), blob.size(),
                     hipMemcpyHostToDevice));
    HIP_OK(hipMalloc(&zero_page, 512 * sizeof(unsigned short)));
    HIP_OK(hipMemset(zero_page, 0, 512 * sizeof(unsigned short)));
    in_numel = 1;
    for (long s : m.in_shape) in_numel *= s;
    HIP_OK(hipMalloc(&input, in_numel * sizeof(unsigned short)));
    long max_pool = 0;
    bool has_softmax = false;
    for (auto& op : m.ops) {
      if (op.kind == "pool") max_pool = std::max(max_pool, op.out_numel);
      if (op.kind == "softmax") has_softmax = true;
    }
    if (max_pool) HIP_OK(hipMalloc(&offs, max_pool));
    if (has_softmax) HIP_OK(hipMalloc(&max_idx, m.batch * sizeof(int)));
    // one output buffer per producing op (inference footprint is tiny
    // next to 288 GB; keeps softmax/act in-place semantics trivial)
    outs.resize(m.ops.size(), nullptr);
    for (size_t i = 0; i < m.ops.size(); ++i) {
      const Op& op = m.ops[i];
      if (op.kind == "softmax" || op.kind == "act") continue;
      size_t elt = (op.kind == "fc" && op.out_f32) ? 4 : 2;
      HIP_OK(hipMalloc(&outs[i], op.out_numel * elt));
    }
  }

  void forward() {
    const void* cur = input;
    long cur_numel = in_numel;
    bool cur_f32 = false;
    for (size_t i = 0; i < m.ops.size(); ++i) {
      const Op& op = m.ops[i];
      if (op.kind == "fc") {
        gemm_nt_dense((const unsigned short*)cur,
                      (const unsigned short*)tptr(op.w), outs[i],
                      op.out_f32, (const float*)tptr(op.bias), zero_page,
                      op.M, op.N, op.K, op.act, stream);
        cur = outs[i];
        cur_numel = op.out_numel;
        cur_f32 = op.out_f32;
      } else if (op.kind == "softmax") {
        softmax_launch((float*)cur, max_idx, op.rows, op.cols, stream);
      } else if (op.kind == "conv") {
        ConvDesc d{op.desc[0], op.desc[1], op.desc[2],  op.desc[3],
                   op.desc[4], op.desc[5], op.desc[6],  op.desc[7],
                   op.desc[8], op.desc[9], op.desc[10], op.desc[11],
                   op.desc[12]};
        finish_conv_desc(d);
        conv_fwd_gemm((const unsigned short*)cur,
                      (const unsigned short*)tptr(op.w), outs[i], false,
                      (const float*)tptr(op.bias), zero_page, d, op.act,
                      stream);
        cur = outs[i];
        cur_numel = op.out_numel;
        cur_f32 = false;
      } else if (op.kind == "pool") {
        PoolDesc d{op.desc[0], op.desc[1], op.desc[2], op.desc[3],
                   op.desc[4], op.desc[5], op.desc[6], op.desc[7],
                   op.desc[8], op.desc[9], 0ull};
        pool_fwd_launch((const unsigned short*)cur,
                        (unsigned short*)outs[i], offs, d, op.mode,
                        stream);
        cur = outs[i];
        cur_numel = op.out_numel;
        cur_f32 = false;
      } else if (op.kind == "lrn") {
        lrn_fwd_launch((const unsigned short*)cur,
                       (unsigned short*)outs[i], nullptr, op.lrn_rows,
                       op.C, op.alpha, op.beta, op.k, op.win, stream);
        cur = outs[i];
        cur_numel = op.out_numel;
        cur_f32 = false;
      } else if (op.kind == "act") {
        act_fwd_launch((unsigned short*)const_cast<void*>(cur),
                       cur_numel, op.act, stream);
      }
    }
    final_buf = const_cast<void*>(cur);
    final_f32 = cur_f32;
    final_numel = cur_numel;
  }
};

}  // namespace

int main(int argc, char** argv) {
  if (argc < 2) {
    std::fprintf(stderr,
                 "usage: mzserve <export_dir> [--input f32.bin] "
                 "[--out f32.bin] [--batches K] [--warmup W] "
                 "[--no-graph] [--parse-only] [--h2d]\n");
    return 2;
  }
  std::string dir = argv[1], in_file, out_file;
  int batches = 20, warmup = 3;
  bool use_graph = true, parse_only = false, time_h2d = false;
  for (int i = 2; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "--input" && i + 1 < argc) in_file = argv[++i];
    else if (a == "--out" && i + 1 < argc) out_file = argv[++i];
    else if (a == "--batches" && i + 1 < argc) batches = atoi(argv[++i]);
    else if (a == "--warmup" && i + 1 < argc) warmup = atoi(argv[++i]);
    else if (a == "--no-graph") use_graph = false;
    else if (a == "--parse-only") parse_only = true;
    else if (a == "--h2d") time_h2d = true;  // re-upload per batch
  }

  Engine eng;
  eng.m = parse_manifest(dir);
  if (parse_only) {
    std::printf("{\"ops\": %zu, \"tensors\": %zu, \"batch\": %d}\n",
                eng.m.ops.size(), eng.m.tensors.size(), eng.m.batch);
    return 0;
  }
  eng.setup(dir);

  // input: fp32 file (one batch, reused) or deterministic random
  std::vector<float> host_in(eng.in_numel);
  if (!in_file.empty()) {
    std::ifstream f(in_file, std::ios::binary);
    f.read((char*)host_in.data(), eng.in_numel * 4);
    if (f.gcount() != (std::streamsize)(eng.in_numel * 4)) {
      std::fprintf(stderr, "mzserve: %s: need %ld fp32 values\n",
                   in_file.c_str(), eng.in_numel);
      return 1;
    }
  } else {
    std::mt19937 rng(12345);
    std::uniform_real_distribution<float> u(-1.f, 1.f);
    for (float& v : host_in) v = u(rng);
  }
  // pinned staging in the device dtype (mirrors the Python engine's
  // convert-then-upload host path)
  unsigned short* host_bf = nullptr;
  HIP_OK(hipHostMalloc(&host_bf, eng.in_numel * 2));
  for (long i = 0; i < eng.in_numel; ++i) host_bf[i] = f2bf(host_in[i]);
  HIP_OK(hipMemcpyAsync(eng.input, host_bf, eng.in_numel * 2,
                        hipMemcpyHostToDevice, eng.stream));

  // warmup (also allocates the FC split-K workspace pre-capture)
  for (int i = 0; i < std::max(warmup, 1); ++i) eng.forward();
  HIP_OK(hipStreamSynchronize(eng.stream));

  hipGraphExec_t gexec = nullptr;
  if (use_graph) {
    hipGraph_t graph = nullptr;
    if (hipStreamBeginCapture(eng.stream, hipStreamCaptureModeGlobal) ==
        hipSuccess) {
      eng.forward();
      if (hipStreamEndCapture(eng.stream, &graph) == hipSuccess &&
          hipGraphInstantiate(&gexec, graph, nullptr, nullptr, 0) ==
              hipSuccess) {
        // replay the captured batch once: capture records, doesn't run
        HIP_OK(hipGraphLaunch(gexec, eng.stream));
        HIP_OK(hipStreamSynchronize(eng.stream));
      } else {
        gexec = nullptr;  // eager fallback
      }
    }
  }

  auto t0 = std::chrono::steady_clock::now();
  for (int i = 0; i < batches; ++i) {
    if (time_h2d) {
      // end-to-end mode: fresh fp32→bf16 convert + pinned upload per
      // batch, like a real request path
      for (long j = 0; j < eng.in_numel; ++j)
        host_bf[j] = f2bf(host_in[j]);
      HIP_OK(hipMemcpyAsync(eng.input, host_bf, eng.in_numel * 2,
                            hipMemcpyHostToDevice, eng.stream));
    }
    if (gexec) HIP_OK(hipGraphLaunch(gexec, eng.stream));
    else eng.forward();
    if (time_h2d) HIP_OK(hipStreamSynchronize(eng.stream));
  }
  HIP_OK(hipStreamSynchronize(eng.stream));
  auto t1 = std::chrono::steady_clock::now();
  double ms = std::chrono::duration<double, std::milli>(t1 - t0).count() /
              batches;

  if (!out_file.empty()) {
    std::vector<float> out(eng.final_numel);
    if (eng.final_f32) {
      HIP_OK(hipMemcpy(out.data(), eng.final_buf, eng.final_numel * 4,
                       hipMemcpyDeviceToHost));
    } else {
      std::vector<unsigned short> raw(eng.final_numel);
      HIP_OK(hipMemcpy(raw.data(), eng.final_buf, eng.final_numel * 2,
                       hipMemcpyDeviceToHost));
      for (long i = 0; i < eng.final_numel; ++i) out[i] = bf2f(raw[i]);
    }
    std::ofstream f(out_file, std::ios::binary);
    f.write((const char*)out.data(), eng.final_numel * 4);
  }

  std::printf(
      "{\"engine\": \"mzserve\", \"batch\": %d, \"batches\": %d, "
      "\"warmup\": %d, \"ms_per_batch\": %.4f, \"img_s\": %.1f, "
      "\"graph\": %s, \"ops\": %zu, \"h2d\": %s}\n",
      eng.m.batch, batches, warmup, ms, eng.m.batch * 1000.0 / ms,
      gexec ? "true" : "false", eng.m.ops.size(),
      time_h2d ? "true" : "false");
  return 0;
}
