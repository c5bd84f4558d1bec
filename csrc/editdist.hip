// Levenshtein edit distance on gfx950 (ref capability:
// /root/reference/src/daft-functions-utf8/src/levenshtein.rs — CPU DP;
// here one THREAD per row pair, rolling DP row in a global scratch
// buffer).  ASCII fast path: rows containing multi-byte UTF-8 report -1
// and fall back to the host implementation (distance is defined over
// characters, which equals bytes only for ASCII).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using torch::Tensor;

namespace {

__global__ void levenshtein_kernel(const int64_t* __restrict__ a_off,
                                   const uint8_t* __restrict__ a_bytes,
                                   const int64_t* __restrict__ b_off,
                                   const uint8_t* __restrict__ b_bytes,
                                   int32_t* __restrict__ scratch,
                                   int64_t scratch_stride, int64_t n,
                                   int32_t max_len,
                                   int32_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const int64_t alo = a_off[i], ahi = a_off[i + 1];
    const int64_t blo = b_off[i], bhi = b_off[i + 1];
    const int la = (int)(ahi - alo), lb = (int)(bhi - blo);
    if (la > max_len || lb > max_len) {
      out[i] = -1;
      continue;
    }
    bool ascii = true;
    for (int64_t p = alo; p < ahi; ++p)
      if (a_bytes[p] >= 0x80) { ascii = false; break; }
    if (ascii)
      for (int64_t p = blo; p < bhi; ++p)
        if (b_bytes[p] >= 0x80) { ascii = false; break; }
    if (!ascii) {
      out[i] = -1;      // host computes the character-level distance
      continue;
    }
    int32_t* row = scratch + i * scratch_stride;
    for (int j = 0; j <= lb; ++j) row[j] = j;
    for (int x = 1; x <= la; ++x) {
      int32_t prev_diag = row[0];
      row[0] = x;
      const uint8_t ca = a_bytes[alo + x - 1];
      for (int y = 1; y <= lb; ++y) {
        int32_t up = row[y];
        int32_t cost = (ca == b_bytes[blo + y - 1]) ? 0 : 1;
        int32_t v = prev_diag + cost;
        int32_t del = up + 1;
        int32_t ins = row[y - 1] + 1;
        if (del < v) v = del;
        if (ins < v) v = ins;
        row[y] = v;
        prev_diag = up;
      }
    }
    out[i] = row[lb];
  }
}

}  // namespace

Tensor levenshtein(Tensor a_off, Tensor a_bytes, Tensor b_off,
                   Tensor b_bytes, int64_t max_len) {
  int64_t n = a_off.numel() - 1;
  auto dev = a_bytes.device();
  auto out = torch::zeros({std::max<int64_t>(n, 1)},
                          torch::dtype(torch::kInt32).device(dev));
  if (n == 0) return out;
  int64_t stride = max_len + 1;
  // per-row DP scratch; the python layer chunks rows to bound this
  TORCH_CHECK(n * stride * 4 < (4ll << 30),
              "levenshtein scratch too large; chunk the input");
  auto scratch = torch::empty({n * stride},
                              torch::dtype(torch::kInt32).device(dev));
  int block = 256;
  int grid = grid_1d(n, block);
  hipLaunchKernelGGL(levenshtein_kernel, dim3(grid), dim3(block), 0,
                     at::hip::getCurrentHIPStream().stream(),
                     a_off.data_ptr<int64_t>(),
                     (const uint8_t*)a_bytes.data_ptr(),
                     b_off.data_ptr<int64_t>(),
                     (const uint8_t*)b_bytes.data_ptr(),
                     scratch.data_ptr<int32_t>(), stride, n,
                     (int32_t)max_len, out.data_ptr<int32_t>());
  return out;
}
