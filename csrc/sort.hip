// Stable LSD radix argsort on order-encoded u64 keys (gfx950).
// The MI355X-native replacement for the reference's comparison sorts
// (daft-core/src/array/ops/sort.rs): 8 passes x 8-bit digits; per-block LDS
// histograms; stable intra-tile ranking via 64-wide wave ballots; global
// digit offsets composed with torch cumsum on-device (no host sync except
// the single-digit skip check).
//
// Multi-key / string ordering is composed at the Python layer by repeated
// stable passes (kernels/rowops.py argsort_multi), so this kernel only ever
// sorts u64.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"
#include "api.h"

static hipStream_t sort_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

constexpr int RADIX = 256;
constexpr int BLOCK = 256;
constexpr int NWAVES = BLOCK / WAVE;

__global__ void radix_hist_kernel(const uint64_t* keys, int64_t n,
                                  int64_t chunk, int shift,
                                  int32_t* hist /*[nblocks][RADIX]*/) {
  __shared__ int32_t lhist[RADIX];
  for (int d = threadIdx.x; d < RADIX; d += blockDim.x) lhist[d] = 0;
  __syncthreads();
  int64_t begin = (int64_t)blockIdx.x * chunk;
  int64_t end = min(begin + chunk, n);
  for (int64_t i = begin + threadIdx.x; i < end; i += blockDim.x) {
    int d = (int)((keys[i] >> shift) & 0xFF);
    atomicAdd(&lhist[d], 1);
  }
  __syncthreads();
  for (int d = threadIdx.x; d < RADIX; d += blockDim.x)
    hist[(int64_t)blockIdx.x * RADIX + d] = lhist[d];
}

// Stable scatter: block b owns input range [b*chunk, (b+1)*chunk), processed
// as sequential 256-element tiles.  Within a tile, each lane finds its rank
// among equal digits via 8 ballot rounds; cross-wave offsets via LDS.
__global__ void radix_scatter_kernel(const uint64_t* keys_in,
                                     const int64_t* vals_in, int64_t n,
                                     int64_t chunk, int shift,
                                     const int32_t* offsets /*[nb][RADIX]*/,
                                     uint64_t* keys_out, int64_t* vals_out) {
  __shared__ int32_t running[RADIX];       // digit offset consumed so far
  __shared__ int32_t wave_hist[NWAVES][RADIX];
  for (int d = threadIdx.x; d < RADIX; d += blockDim.x)
    running[d] = offsets[(int64_t)blockIdx.x * RADIX + d];
  int64_t begin = (int64_t)blockIdx.x * chunk;
  int64_t end = min(begin + chunk, n);
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  uint64_t lane_lt = (lane == 63) ? ~0ull >> 1 : ((1ull << lane) - 1);
  for (int64_t tile = begin; tile < end; tile += BLOCK) {
    for (int d = threadIdx.x; d < RADIX; d += blockDim.x)
      for (int w = 0; w < NWAVES; ++w) wave_hist[w][d] = 0;
    __syncthreads();
    int64_t i = tile + threadIdx.x;
    bool active = i < end;
    uint64_t k = active ? keys_in[i] : 0;
    int d = (int)((k >> shift) & 0xFF);
    // peers: lanes in this wave with the same digit (among active lanes)
    uint64_t peers = __ballot(active);
    for (int b = 0; b < 8; ++b) {
      bool bit = (d >> b) & 1;
      uint64_t vote = __ballot(bit);
      peers &= bit ? vote : ~vote;
    }
    int rank_in_wave = __popcll(peers & lane_lt);
    int wave_count = __popcll(peers);
    if (active && rank_in_wave == 0) wave_hist[wid][d] = wave_count;
    __syncthreads();
    if (active) {
      int32_t before = 0;
      for (int w = 0; w < wid; ++w) before += wave_hist[w][d];
      int64_t pos = running[d] + before + rank_in_wave;
      keys_out[pos] = k;
      vals_out[pos] = vals_in[i];
    }
    __syncthreads();
    for (int dd = threadIdx.x; dd < RADIX; dd += blockDim.x) {
      int32_t t = 0;
      for (int w = 0; w < NWAVES; ++w) t += wave_hist[w][dd];
      running[dd] += t;
    }
    __syncthreads();
  }
}

Tensor radix_argsort(Tensor keys) {
  TORCH_CHECK(keys.is_cuda() && keys.dtype() == torch::kInt64);
  keys = keys.contiguous();
  auto dev = keys.device();
  int64_t n = keys.numel();
  auto opts64 = torch::dtype(torch::kInt64).device(dev);
  auto vals = torch::arange(n, opts64);
  if (n <= 1) return vals;

  int64_t chunk = (n + kMaxBlocks - 1) / kMaxBlocks;
  if (chunk < BLOCK) chunk = BLOCK;
  int nblocks = (int)((n + chunk - 1) / chunk);

  auto keys_a = keys.clone();
  auto keys_b = torch::empty_like(keys_a);
  auto vals_a = vals;
  auto vals_b = torch::empty_like(vals_a);
  auto hist = torch::empty({(int64_t)nblocks, RADIX},
                           torch::dtype(torch::kInt32).device(dev));

  for (int pass = 0; pass < 8; ++pass) {
    int shift = pass * 8;
    hipLaunchKernelGGL(radix_hist_kernel, dim3(nblocks), dim3(BLOCK), 0,
                       sort_stream(),
                       (const uint64_t*)keys_a.data_ptr<int64_t>(), n, chunk,
                       shift, hist.data_ptr<int32_t>());
    auto col_totals = hist.sum(0);  // [RADIX] int64
    // skip-pass: single populated digit => already grouped for this byte
    auto nz = (col_totals > 0).sum().item<int64_t>();
    if (nz <= 1) continue;
    auto digit_base = torch::cumsum(col_totals, 0) - col_totals;  // excl
    auto block_excl = torch::cumsum(hist.to(torch::kInt64), 0) -
                      hist.to(torch::kInt64);
    auto offsets = (digit_base.unsqueeze(0) + block_excl).to(torch::kInt32)
                       .contiguous();
    hipLaunchKernelGGL(radix_scatter_kernel, dim3(nblocks), dim3(BLOCK), 0,
                       sort_stream(),
                       (const uint64_t*)keys_a.data_ptr<int64_t>(),
                       vals_a.data_ptr<int64_t>(), n, chunk, shift,
                       offsets.data_ptr<int32_t>(),
                       (uint64_t*)keys_b.data_ptr<int64_t>(),
                       vals_b.data_ptr<int64_t>());
    std::swap(keys_a, keys_b);
    std::swap(vals_a, vals_b);
  }
  return vals_a;
}

// big-endian 8-byte chunk key for lexicographic string sorting
__global__ void string_chunk_key_kernel(const int64_t* offs,
                                        const uint8_t* bytes, int64_t n,
                                        int64_t base, uint64_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t a = offs[i] + base, b = offs[i + 1];
    uint64_t k = 0;
    if (a < b) {
      int m = (int)min((int64_t)8, b - a);
      for (int j = 0; j < m; ++j)
        k |= (uint64_t)bytes[a + j] << (56 - 8 * j);
    }
    out[i] = k;
  }
}

Tensor string_chunk_key(Tensor offsets, Tensor bytes, int64_t chunk) {
  auto dev = offsets.device();
  int64_t n = offsets.numel() - 1;
  auto out = torch::zeros({n}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) {
    int block = 256;
    const uint8_t* bp = bytes.numel() ? bytes.data_ptr<uint8_t>() : nullptr;
    hipLaunchKernelGGL(string_chunk_key_kernel, dim3(grid_1d(n, block)),
                       dim3(block), 0, sort_stream(),
                       offsets.data_ptr<int64_t>(), bp, n, chunk * 8,
                       (uint64_t*)out.data_ptr<int64_t>());
  }
  return out;
}
