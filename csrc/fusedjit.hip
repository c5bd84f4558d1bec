// hipRTC-backed JIT for fused expression kernels (gfx950).
//
// The generic interpreter (fusedexpr.hip) pays for its dynamically
// indexed stack with scratch traffic; a projection list compiled to
// straight-line code runs at memory bandwidth.  Python generates the
// kernel source per (expression list, schema, validity pattern) and this
// launcher compiles it ONCE per process (cache keyed by source) with
// hipRTC, then launches through the module API.  Column/output buffers
// travel via the same packed descriptor arrays as the interpreter, so a
// cached kernel is reusable across batches and steps.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hiprtc.h>

#include <mutex>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

using torch::Tensor;
using OptTensor = c10::optional<Tensor>;

namespace {

struct JitEntry {
  hipModule_t mod = nullptr;
  hipFunction_t fn = nullptr;
};

std::unordered_map<std::string, JitEntry>& jit_cache() {
  static std::unordered_map<std::string, JitEntry> c;
  return c;
}
std::mutex g_jit_mu;

hipFunction_t compile_or_get(const std::string& src) {
  std::lock_guard<std::mutex> lk(g_jit_mu);
  auto& cache = jit_cache();
  auto it = cache.find(src);
  if (it != cache.end()) {
    if (it->second.fn == nullptr)
      throw std::runtime_error("fused jit: compile previously failed");
    return it->second.fn;
  }

  hiprtcProgram prog;
  if (hiprtcCreateProgram(&prog, src.c_str(), "fe.hip", 0, nullptr,
                          nullptr) != HIPRTC_SUCCESS)
    throw std::runtime_error("hiprtcCreateProgram failed");
  const char* opts[] = {"--offload-arch=gfx950", "-O3"};
  hiprtcResult rc = hiprtcCompileProgram(prog, 2, opts);
  if (rc != HIPRTC_SUCCESS) {
    size_t log_size = 0;
    hiprtcGetProgramLogSize(prog, &log_size);
    std::string log(log_size, '\0');
    hiprtcGetProgramLog(prog, log.data());
    hiprtcDestroyProgram(&prog);
    // cache the failure: retrying an impossible compile per batch would
    // cost ~150 ms every call
    cache.emplace(src, JitEntry{});
    throw std::runtime_error("hiprtc compile failed:\n" + log);
  }
  size_t code_size = 0;
  hiprtcGetCodeSize(prog, &code_size);
  std::vector<char> code(code_size);
  hiprtcGetCode(prog, code.data());
  hiprtcDestroyProgram(&prog);

  JitEntry e;
  if (hipModuleLoadData(&e.mod, code.data()) != hipSuccess)
    throw std::runtime_error("hipModuleLoadData failed");
  if (hipModuleGetFunction(&e.fn, e.mod, "fe") != hipSuccess)
    throw std::runtime_error("hipModuleGetFunction(fe) failed");
  cache.emplace(src, e);
  return e.fn;
}

}  // namespace

bool fused_jit_available() {
  return true;
}

std::vector<Tensor> fused_eval_jit(const std::string& src,
                                   std::vector<Tensor> cols,
                                   std::vector<OptTensor> valids,
                                   std::vector<int64_t> out_codes,
                                   std::vector<int64_t> out_need_valid,
                                   int64_t n) {
  auto dev = cols.empty() ? torch::Device(torch::kCUDA, 0)
                          : cols[0].device();
  int ncols = (int)cols.size();
  int nouts = (int)out_codes.size();

  // pack column descriptors {data, valid} (dtype/scale are baked into the
  // generated source)
  auto hc = torch::empty({std::max(ncols, 1) * 2},
                         torch::dtype(torch::kInt64));
  int64_t* h = hc.data_ptr<int64_t>();
  for (int i = 0; i < ncols; ++i) {
    h[i * 2 + 0] = (int64_t)cols[i].data_ptr();
    h[i * 2 + 1] =
        valids[i].has_value() ? (int64_t)valids[i]->data_ptr<bool>() : 0;
  }
  auto dcols = hc.to(dev);

  static const torch::ScalarType kOutTy[] = {
      torch::kFloat64, torch::kFloat32, torch::kInt64,
      torch::kInt32,   torch::kInt16,   torch::kInt8,
      torch::kBool};
  std::vector<Tensor> results;
  auto ho = torch::empty({std::max(nouts, 1) * 2},
                         torch::dtype(torch::kInt64));
  int64_t* ho_p = ho.data_ptr<int64_t>();
  for (int k = 0; k < nouts; ++k) {
    auto out = torch::empty({n},
                            torch::dtype(kOutTy[out_codes[k]]).device(dev));
    Tensor vout;
    bool nv = out_need_valid[k] != 0;
    if (nv) vout = torch::empty({n}, torch::dtype(torch::kBool).device(dev));
    ho_p[k * 2 + 0] = (int64_t)out.data_ptr();
    ho_p[k * 2 + 1] = nv ? (int64_t)vout.data_ptr<bool>() : 0;
    results.push_back(out);
    results.push_back(nv ? vout : Tensor());
  }
  auto douts = ho.to(dev);
  if (n == 0) return results;

  hipFunction_t fn = compile_or_get(src);
  void* cols_p = (void*)dcols.data_ptr();
  void* outs_p = (void*)douts.data_ptr();
  long long nn = (long long)n;
  void* params[] = {&cols_p, &outs_p, &nn};
  int block = 256;
  int grid = grid_1d(n, block, /*items_per_thread=*/2);
  hipStream_t stream = at::hip::getCurrentHIPStream().stream();
  hipError_t rc = hipModuleLaunchKernel(fn, grid, 1, 1, block, 1, 1, 0,
                                        stream, params, nullptr);
  if (rc != hipSuccess)
    throw std::runtime_error(std::string("fused jit launch failed: ") +
                             hipGetErrorString(rc));
  return results;
}
