#include "hip/hip_runtime.h"
// String kernels over Arrow offsets+bytes (gfx950) — the GPU equivalents of
// the reference's daft-functions-utf8 kernels.  One thread per row for
// predicate/search ops (TPC-H string rows are short); byte-parallel for
// case mapping.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"
#include "api.h"

static hipStream_t str_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// mode: 0 = find (first match position, -1 if none)
//       1 = prefix test (0 or -1)
//       2 = suffix test (0 or -1)
__global__ void str_find_kernel(const int64_t* offs, const uint8_t* bytes,
                                int64_t n, const uint8_t* pat, int64_t plen,
                                int mode, int64_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t a = offs[i], b = offs[i + 1];
    int64_t len = b - a;
    const uint8_t* s = bytes + a;
    int64_t res = -1;
    if (plen == 0) {
      res = 0;
    } else if (len >= plen) {
      if (mode == 1) {
        bool ok = true;
        for (int64_t j = 0; j < plen; ++j)
          if (s[j] != pat[j]) { ok = false; break; }
        res = ok ? 0 : -1;
      } else if (mode == 2) {
        bool ok = true;
        const uint8_t* t = s + len - plen;
        for (int64_t j = 0; j < plen; ++j)
          if (t[j] != pat[j]) { ok = false; break; }
        res = ok ? 0 : -1;
      } else {
        for (int64_t p = 0; p + plen <= len; ++p) {
          if (s[p] != pat[0]) continue;
          bool ok = true;
          for (int64_t j = 1; j < plen; ++j)
            if (s[p + j] != pat[j]) { ok = false; break; }
          if (ok) { res = p; break; }
        }
      }
    }
    out[i] = res;
  }
}

Tensor str_find(Tensor offsets, Tensor bytes, Tensor pattern, int64_t mode) {
  auto dev = offsets.device();
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) {
    int block = 256;
    const uint8_t* bp = bytes.numel() ? bytes.data_ptr<uint8_t>() : nullptr;
    const uint8_t* pp = pattern.numel() ? pattern.data_ptr<uint8_t>() : nullptr;
    hipLaunchKernelGGL(str_find_kernel, dim3(grid_1d(n, block)), dim3(block),
                       0, str_stream(), offsets.data_ptr<int64_t>(), bp, n,
                       pp, pattern.numel(), (int)mode,
                       out.data_ptr<int64_t>());
  }
  return out;
}

// ordered multi-substring LIKE (patterns without '_'): needles concatenated
// in `blob` with lengths in `lens`; anchored prefix/suffix flags.
__global__ void str_like_kernel(const int64_t* offs, const uint8_t* bytes,
                                int64_t n, const uint8_t* blob,
                                const int64_t* lens, int nneedles,
                                bool anchored_prefix, bool anchored_suffix,
                                bool* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t a = offs[i], b = offs[i + 1];
    int64_t len = b - a;
    const uint8_t* s = bytes + a;
    bool ok = true;
    int64_t pos = 0;
    int64_t blob_off = 0;
    for (int k = 0; k < nneedles && ok; ++k) {
      const uint8_t* nd = blob + blob_off;
      int64_t nl = lens[k];
      blob_off += nl;
      bool is_first = (k == 0), is_last = (k == nneedles - 1);
      if (is_first && anchored_prefix) {
        if (len < nl) { ok = false; break; }
        for (int64_t j = 0; j < nl; ++j)
          if (s[j] != nd[j]) { ok = false; break; }
        pos = nl;
        continue;
      }
      if (is_last && anchored_suffix) {
        if (len - pos < nl) { ok = false; break; }
        const uint8_t* t = s + len - nl;
        for (int64_t j = 0; j < nl; ++j)
          if (t[j] != nd[j]) { ok = false; break; }
        continue;
      }
      // search nd in s[pos..]
      int64_t found = -1;
      for (int64_t p = pos; p + nl <= len; ++p) {
        if (s[p] != nd[0]) continue;
        bool m = true;
        for (int64_t j = 1; j < nl; ++j)
          if (s[p + j] != nd[j]) { m = false; break; }
        if (m) { found = p; break; }
      }
      if (found < 0) { ok = false; break; }
      pos = found + nl;
    }
    out[i] = ok;
  }
}

Tensor str_like(Tensor offsets, Tensor bytes, Tensor needles, Tensor lens,
                bool anchored_prefix, bool anchored_suffix) {
  auto dev = offsets.device();
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, torch::dtype(torch::kBool).device(dev));
  if (n > 0) {
    int block = 256;
    const uint8_t* bp = bytes.numel() ? bytes.data_ptr<uint8_t>() : nullptr;
    const uint8_t* np = needles.numel() ? needles.data_ptr<uint8_t>() : nullptr;
    hipLaunchKernelGGL(str_like_kernel, dim3(grid_1d(n, block)), dim3(block),
                       0, str_stream(), offsets.data_ptr<int64_t>(), bp, n,
                       np, lens.data_ptr<int64_t>(), (int)lens.numel(),
                       anchored_prefix, anchored_suffix,
                       out.data_ptr<bool>());
  }
  return out;
}

// ASCII case mapping, byte-parallel (UTF-8 multibyte left untouched).
__global__ void str_case_kernel(const uint8_t* in, int64_t nbytes, int mode,
                                uint8_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nbytes;
       i += stride) {
    uint8_t c = in[i];
    if (mode == 0) {  // lower
      if (c >= 'A' && c <= 'Z') c += 32;
    } else {
      if (c >= 'a' && c <= 'z') c -= 32;
    }
    out[i] = c;
  }
}

Tensor str_case(Tensor offsets, Tensor bytes, int64_t mode) {
  auto out = torch::empty_like(bytes);
  int64_t nb = bytes.numel();
  if (nb > 0) {
    int block = 256;
    hipLaunchKernelGGL(str_case_kernel, dim3(grid_1d(nb, block)),
                       dim3(block), 0, str_stream(),
                       bytes.data_ptr<uint8_t>(), nb, (int)mode,
                       out.data_ptr<uint8_t>());
  }
  return out;
}

__global__ void substr_copy_kernel(const int64_t* src_off,
                                   const uint8_t* src_bytes, int64_t start,
                                   const int64_t* out_off, int64_t n,
                                   int64_t total, uint8_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < total;
       j += stride) {
    // binary search row
    int64_t lo = 0, hi = n;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      if (out_off[mid + 1] <= j) lo = mid + 1; else hi = mid;
    }
    int64_t within = j - out_off[lo];
    out[j] = src_bytes[src_off[lo] + start + within];
  }
}

std::vector<Tensor> str_substr(Tensor offsets, Tensor bytes, int64_t start,
                               int64_t length) {
  auto dev = offsets.device();
  int64_t n = offsets.numel() - 1;
  auto lens = offsets.slice(0, 1) - offsets.slice(0, 0, n);
  auto new_lens = (lens - start).clamp(0);
  if (length >= 0) new_lens = new_lens.clamp(0, length);
  auto out_off = torch::zeros({n + 1}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) out_off.slice(0, 1).copy_(torch::cumsum(new_lens, 0));
  int64_t total = n > 0 ? out_off[n].item<int64_t>() : 0;
  auto out = torch::empty({total}, torch::dtype(torch::kUInt8).device(dev));
  if (total > 0) {
    int block = 256;
    hipLaunchKernelGGL(substr_copy_kernel, dim3(grid_1d(total, block)),
                       dim3(block), 0, str_stream(),
                       offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
                       start, out_off.data_ptr<int64_t>(), n, total,
                       out.data_ptr<uint8_t>());
  }
  return {out_off, out};
}

// row-wise concat of K string columns
__global__ void str_concat_kernel(const int64_t* const* offs,
                                  const uint8_t* const* datas, int k,
                                  const int64_t* out_off, int64_t n,
                                  uint8_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t pos = out_off[i];
    for (int c = 0; c < k; ++c) {
      int64_t a = offs[c][i], b = offs[c][i + 1];
      const uint8_t* s = datas[c] + a;
      for (int64_t j = 0; j < b - a; ++j) out[pos++] = s[j];
    }
  }
}

std::vector<Tensor> str_concat(const std::vector<Tensor>& offsets,
                               const std::vector<Tensor>& bytes) {
  auto dev = offsets[0].device();
  int64_t n = offsets[0].numel() - 1;
  int k = (int)offsets.size();
  auto lens = torch::zeros({n}, torch::dtype(torch::kInt64).device(dev));
  for (int c = 0; c < k; ++c)
    lens += offsets[c].slice(0, 1) - offsets[c].slice(0, 0, n);
  auto out_off = torch::zeros({n + 1}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) out_off.slice(0, 1).copy_(torch::cumsum(lens, 0));
  int64_t total = n > 0 ? out_off[n].item<int64_t>() : 0;
  auto out = torch::empty({total}, torch::dtype(torch::kUInt8).device(dev));
  if (total > 0) {
    // pointer arrays on device
    auto hoffs = torch::empty({k}, torch::dtype(torch::kInt64));
    auto hdata = torch::empty({k}, torch::dtype(torch::kInt64));
    for (int c = 0; c < k; ++c) {
      hoffs[c] = (int64_t)offsets[c].data_ptr<int64_t>();
      hdata[c] = (int64_t)(bytes[c].numel() ? bytes[c].data_ptr<uint8_t>()
                                            : nullptr);
    }
    auto doffs = hoffs.to(dev);
    auto ddata = hdata.to(dev);
    int block = 256;
    hipLaunchKernelGGL(str_concat_kernel, dim3(grid_1d(n, block)),
                       dim3(block), 0, str_stream(),
                       (const int64_t* const*)doffs.data_ptr<int64_t>(),
                       (const uint8_t* const*)ddata.data_ptr<int64_t>(), k,
                       out_off.data_ptr<int64_t>(), n,
                       out.data_ptr<uint8_t>());
  }
  return {out_off, out};
}

__global__ void char_length_kernel(const int64_t* offs, const uint8_t* bytes,
                                   int64_t n, int64_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t a = offs[i], b = offs[i + 1];
    int64_t cnt = 0;
    for (int64_t j = a; j < b; ++j)
      cnt += (bytes[j] & 0xC0) != 0x80;  // count non-continuation bytes
    out[i] = cnt;
  }
}

Tensor str_char_length(Tensor offsets, Tensor bytes) {
  auto dev = offsets.device();
  int64_t n = offsets.numel() - 1;
  auto out = torch::zeros({n}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) {
    int block = 256;
    const uint8_t* bp = bytes.numel() ? bytes.data_ptr<uint8_t>() : nullptr;
    hipLaunchKernelGGL(char_length_kernel, dim3(grid_1d(n, block)),
                       dim3(block), 0, str_stream(),
                       offsets.data_ptr<int64_t>(), bp, n,
                       out.data_ptr<int64_t>());
  }
  return out;
}

__global__ void string_compare_kernel(const int64_t* ao, const uint8_t* ab,
                                      const int64_t* bo, const uint8_t* bb,
                                      int64_t n, int8_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t sa = ao[i], ea = ao[i + 1];
    int64_t sb = bo[i], eb = bo[i + 1];
    int64_t la = ea - sa, lb = eb - sb;
    int64_t m = min(la, lb);
    int8_t r = 0;
    for (int64_t j = 0; j < m; ++j) {
      uint8_t x = ab[sa + j], y = bb[sb + j];
      if (x != y) { r = x < y ? -1 : 1; break; }
    }
    if (r == 0 && la != lb) r = la < lb ? -1 : 1;
    out[i] = r;
  }
}

Tensor string_compare(Tensor a_off, Tensor a_bytes, Tensor b_off,
                      Tensor b_bytes) {
  auto dev = a_off.device();
  int64_t n = a_off.numel() - 1;
  auto out = torch::zeros({n}, torch::dtype(torch::kInt8).device(dev));
  if (n > 0) {
    int block = 256;
    const uint8_t* ap = a_bytes.numel() ? a_bytes.data_ptr<uint8_t>() : nullptr;
    const uint8_t* bp = b_bytes.numel() ? b_bytes.data_ptr<uint8_t>() : nullptr;
    hipLaunchKernelGGL(string_compare_kernel, dim3(grid_1d(n, block)),
                       dim3(block), 0, str_stream(),
                       a_off.data_ptr<int64_t>(), ap,
                       b_off.data_ptr<int64_t>(), bp, n,
                       out.data_ptr<int8_t>());
  }
  return out;
}
