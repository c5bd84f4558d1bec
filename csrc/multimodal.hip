// Multimodal kernels (gfx950): MinHash signatures, HyperLogLog updates,
// bilinear image resize.  Ref capabilities: daft-minhash (portable_simd
// minhash, src/lib.rs:279-330), hyperloglog (2^14 dense registers), and
// daft-image resize (src/series.rs:123).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"
#include "api.h"

static hipStream_t mm_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
// MinHash: one wave per row.  Every lane scans the row's word n-grams
// (recomputing the cheap ngram hash locally — no cross-lane traffic), and
// keeps the running min for its subset of permutations in registers:
// lane l owns permutations l, l+64, l+128, ...
// perm_i(h) = (a_i * h + b_i) mod (2^61 - 1), truncated to u32
// ---------------------------------------------------------------------------

constexpr uint64_t MERSENNE61 = (1ull << 61) - 1;
constexpr int MAX_PERMS_PER_LANE = 8;  // up to 512 permutations

DEV_INLINE uint64_t mul_mod61(uint64_t a, uint64_t b) {
  // 61-bit modular multiply via 128-bit product
  __uint128_t p = (__uint128_t)a * b;
  uint64_t lo = (uint64_t)(p & MERSENNE61);
  uint64_t hi = (uint64_t)(p >> 61);
  uint64_t s = lo + hi;
  if (s >= MERSENNE61) s -= MERSENNE61;
  return s;
}

__global__ void minhash_kernel(const int64_t* offs, const uint8_t* bytes,
                               int64_t n, int num_hashes, int ngram_size,
                               const uint64_t* pa, const uint64_t* pb,
                               uint32_t* out /* [n, num_hashes] */) {
  int wave_in_block = threadIdx.x / WAVE;
  int lane = threadIdx.x & (WAVE - 1);
  int waves_per_block = blockDim.x / WAVE;
  int64_t row0 = (int64_t)blockIdx.x * waves_per_block + wave_in_block;
  int64_t row_stride = (int64_t)gridDim.x * waves_per_block;

  int nper = (num_hashes + WAVE - 1) / WAVE;
  uint64_t mins[MAX_PERMS_PER_LANE];

  for (int64_t row = row0; row < n; row += row_stride) {
    for (int q = 0; q < nper; ++q) mins[q] = ~0ull;
    int64_t a = offs[row], b = offs[row + 1];
    // scan words; maintain a ring of the last `ngram_size` word starts
    int64_t word_starts[16];  // ngram_size <= 16
    int nwords = 0;
    int64_t i = a;
    while (i < b) {
      while (i < b && bytes[i] == ' ') ++i;
      if (i >= b) break;
      int64_t ws = i;
      while (i < b && bytes[i] != ' ') ++i;
      // word is [ws, i)
      word_starts[nwords % 16] = ws;
      ++nwords;
      if (nwords >= ngram_size) {
        int64_t gs = word_starts[(nwords - ngram_size) % 16];
        uint64_t h = hash_bytes_dev(bytes + gs, i - gs);
        h &= MERSENNE61;  // keep within field
        for (int q = 0; q < nper; ++q) {
          int perm = lane + q * WAVE;
          if (perm < num_hashes) {
            uint64_t v = mul_mod61(pa[perm], h) + pb[perm];
            if (v >= MERSENNE61) v -= MERSENNE61;
            if (v < mins[q]) mins[q] = v;
          }
        }
      }
    }
    for (int q = 0; q < nper; ++q) {
      int perm = lane + q * WAVE;
      if (perm < num_hashes)
        out[row * num_hashes + perm] =
            (uint32_t)(mins[q] == ~0ull ? 0xFFFFFFFFu : (mins[q] & 0xFFFFFFFFu));
    }
  }
}

Tensor minhash(Tensor offsets, Tensor bytes, int64_t num_hashes,
               int64_t ngram_size, Tensor perm_a, Tensor perm_b) {
  TORCH_CHECK(ngram_size <= 16, "ngram_size <= 16");
  TORCH_CHECK(num_hashes <= MAX_PERMS_PER_LANE * WAVE,
              "num_hashes <= 512");
  auto dev = offsets.device();
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n * num_hashes},
                          torch::dtype(torch::kInt32).device(dev));
  if (n > 0) {
    int block = 256;
    int waves_per_block = block / WAVE;
    int grid = (int)std::min<int64_t>((n + waves_per_block - 1) /
                                      waves_per_block, kMaxBlocks);
    const uint8_t* bp = bytes.numel() ? bytes.data_ptr<uint8_t>() : nullptr;
    hipLaunchKernelGGL(minhash_kernel, dim3(grid), dim3(block), 0,
                       mm_stream(), offsets.data_ptr<int64_t>(), bp, n,
                       (int)num_hashes, (int)ngram_size,
                       (const uint64_t*)perm_a.data_ptr<int64_t>(),
                       (const uint64_t*)perm_b.data_ptr<int64_t>(),
                       (uint32_t*)out.data_ptr<int32_t>());
  }
  return out;
}

// ---------------------------------------------------------------------------
// HyperLogLog: dense 2^14 u32 registers per group, atomicMax update
// ---------------------------------------------------------------------------

constexpr int HLL_BITS = 14;
constexpr int HLL_REGS = 1 << HLL_BITS;

__global__ void hll_update_kernel(const uint64_t* hashes,
                                  const int64_t* gids, const bool* valid,
                                  int64_t n, uint32_t* regs) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (valid && !valid[i]) continue;
    uint64_t h = hashes[i];
    uint32_t idx = (uint32_t)(h >> (64 - HLL_BITS));
    uint64_t rest = h << HLL_BITS;
    uint32_t rank = rest == 0 ? (64 - HLL_BITS + 1)
                              : (uint32_t)__clzll((long long)rest) + 1;
    int64_t g = gids ? gids[i] : 0;
    atomicMax(&regs[g * HLL_REGS + idx], rank);
  }
}

Tensor hll_update(Tensor hashes, Tensor gids, Tensor valid,
                  int64_t num_groups) {
  auto dev = hashes.device();
  auto regs = torch::zeros({num_groups * HLL_REGS},
                           torch::dtype(torch::kInt32).device(dev));
  int64_t n = hashes.numel();
  if (n > 0) {
    int block = 256;
    hipLaunchKernelGGL(hll_update_kernel, dim3(grid_1d(n, block)),
                       dim3(block), 0, mm_stream(),
                       (const uint64_t*)hashes.data_ptr<int64_t>(),
                       gids.defined() && gids.numel()
                           ? gids.data_ptr<int64_t>() : nullptr,
                       valid.defined() && valid.numel()
                           ? valid.data_ptr<bool>() : nullptr,
                       n, (uint32_t*)regs.data_ptr<int32_t>());
  }
  return regs;
}

// ---------------------------------------------------------------------------
// bilinear image resize: variable-size uint8 HWC inputs (concat bytes +
// per-row dims) -> fixed [H, W, C] outputs.  One thread per output element.
// ---------------------------------------------------------------------------

__global__ void image_resize_kernel(const uint8_t* src,
                                    const int64_t* src_off,
                                    const int32_t* src_h,
                                    const int32_t* src_w, int channels,
                                    int64_t n, int out_h, int out_w,
                                    uint8_t* out) {
  int64_t per_img = (int64_t)out_h * out_w * channels;
  int64_t total = n * per_img;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < total;
       j += stride) {
    int64_t img = j / per_img;
    int64_t rem = j - img * per_img;
    int oy = (int)(rem / (out_w * channels));
    int ox = (int)((rem / channels) % out_w);
    int c = (int)(rem % channels);
    int ih = src_h[img], iw = src_w[img];
    const uint8_t* s = src + src_off[img];
    float fy = (oy + 0.5f) * ih / out_h - 0.5f;
    float fx = (ox + 0.5f) * iw / out_w - 0.5f;
    int y0 = (int)floorf(fy), x0 = (int)floorf(fx);
    float dy = fy - y0, dx = fx - x0;
    int y1 = min(y0 + 1, ih - 1), x1 = min(x0 + 1, iw - 1);
    y0 = max(y0, 0);
    x0 = max(x0, 0);
    float v00 = s[((int64_t)y0 * iw + x0) * channels + c];
    float v01 = s[((int64_t)y0 * iw + x1) * channels + c];
    float v10 = s[((int64_t)y1 * iw + x0) * channels + c];
    float v11 = s[((int64_t)y1 * iw + x1) * channels + c];
    float v = v00 * (1 - dy) * (1 - dx) + v01 * (1 - dy) * dx +
              v10 * dy * (1 - dx) + v11 * dy * dx;
    out[j] = (uint8_t)min(255.0f, max(0.0f, v + 0.5f));
  }
}

Tensor image_resize(Tensor src_bytes, Tensor src_off, Tensor src_h,
                    Tensor src_w, int64_t channels, int64_t out_h,
                    int64_t out_w) {
  auto dev = src_bytes.device();
  int64_t n = src_h.numel();
  auto out = torch::empty({n * out_h * out_w * channels},
                          torch::dtype(torch::kUInt8).device(dev));
  if (n > 0) {
    int block = 256;
    int64_t total = out.numel();
    hipLaunchKernelGGL(image_resize_kernel, dim3(grid_1d(total, block)),
                       dim3(block), 0, mm_stream(),
                       src_bytes.data_ptr<uint8_t>(),
                       src_off.data_ptr<int64_t>(),
                       src_h.data_ptr<int32_t>(), src_w.data_ptr<int32_t>(),
                       (int)channels, n, (int)out_h, (int)out_w,
                       out.data_ptr<uint8_t>());
  }
  return out;
}

// ---------------------------------------------------------------------------
// SimHash: 64-bit near-duplicate fingerprint over byte n-grams (capability
// of /root/reference/src/daft-functions/src/simhash.rs:15-42: per-bit
// majority vote over ngram hashes).  MI355X-native layout: one 64-lane wave
// per row, lane j hashes ngram j (strided); the per-bit vote is 64 wave
// ballots per 64-ngram chunk, and lane i keeps only bit i's popcount —
// no LDS, no cross-lane shuffles beyond ballot.
// ---------------------------------------------------------------------------
__global__ void simhash_kernel(const int64_t* offs, const uint8_t* bytes,
                               int64_t n, int ngram, uint64_t* out) {
  int wave_in_block = threadIdx.x / WAVE;
  int lane = threadIdx.x & (WAVE - 1);
  int waves_per_block = blockDim.x / WAVE;
  int64_t row0 = (int64_t)blockIdx.x * waves_per_block + wave_in_block;
  int64_t row_stride = (int64_t)gridDim.x * waves_per_block;
  for (int64_t row = row0; row < n; row += row_stride) {
    int64_t beg = offs[row], end = offs[row + 1];
    int64_t len = end - beg;
    int64_t m = len - ngram + 1;  // number of ngrams
    if (m <= 0) {
      if (lane == 0) out[row] = 0;
      continue;
    }
    int64_t my_cnt = 0;  // lane i: how many ngrams have bit i set
    for (int64_t base = 0; base < m; base += WAVE) {
      int64_t j = base + lane;
      uint64_t h = 0;
      bool act = j < m;
      if (act) h = hash_bytes_dev(bytes + beg + j, ngram);
      #pragma unroll
      for (int i = 0; i < 64; ++i) {
        uint64_t mask = __ballot(act && ((h >> i) & 1));
        if (lane == i) my_cnt += __popcll(mask);
      }
    }
    // bit i of the fingerprint: weight_i = 2*cnt_i - m > 0
    uint64_t fp = __ballot(2 * my_cnt > m);
    if (lane == 0) out[row] = fp;
  }
}

Tensor simhash(Tensor offsets, Tensor bytes, int64_t ngram_size) {
  auto dev = offsets.device();
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) {
    int block = 256;
    int waves_per_block = block / WAVE;
    int grid = (int)std::min<int64_t>((n + waves_per_block - 1) /
                                      waves_per_block, kMaxBlocks);
    const uint8_t* bp = bytes.numel() ? bytes.data_ptr<uint8_t>() : nullptr;
    hipLaunchKernelGGL(simhash_kernel, dim3(grid), dim3(block), 0,
                       mm_stream(), offsets.data_ptr<int64_t>(), bp, n,
                       (int)ngram_size, (uint64_t*)out.data_ptr<int64_t>());
  }
  return out;
}
