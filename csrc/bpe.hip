// Byte-level BPE encode on gfx950 (ref capability:
// /root/reference/src/daft-functions-tokenize/src/bpe.rs — tiktoken-style
// merges; the reference runs them on CPU threads, here one WAVEFRONT per
// row runs the greedy merge loop with the token sequence in LDS).
//
// Algorithm per row (classic BPE):
//   ids[] = byte2id[bytes]; repeat: find the adjacent pair with the
//   lowest merge rank (wave-parallel scan + shuffle min-reduction),
//   replace it with its merged id (cooperative two-phase LDS shift),
//   until no pair has a rank.
//
// Merge table: open-addressing hash of key=(a<<32)|b -> (rank, new_id),
// built host-side.  Rows longer than BPE_MAX_LEN bytes get count=-1 and
// fall back to the host tokenizer.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using torch::Tensor;

namespace {

constexpr int BPE_MAX_LEN = 4096;
constexpr uint32_t RANK_INF = 0xFFFFFFFFu;

__device__ __forceinline__ void pair_lookup(const uint64_t* keys,
                                            const uint64_t* vals,
                                            int tsize_mask, uint64_t key,
                                            uint32_t* rank,
                                            uint32_t* newid) {
  uint64_t h = key;
  h ^= h >> 33;
  h *= 0xff51afd7ed558ccdull;
  h ^= h >> 29;
  int slot = (int)(h & (uint64_t)tsize_mask);
  for (int probe = 0; probe <= tsize_mask; ++probe) {
    uint64_t k = keys[slot];
    if (k == key) {
      uint64_t v = vals[slot];
      *rank = (uint32_t)(v >> 32);
      *newid = (uint32_t)v;
      return;
    }
    if (k == ~0ull) break;  // empty slot: not present
    slot = (slot + 1) & tsize_mask;
  }
  *rank = RANK_INF;
  *newid = 0;
}

__global__ void bpe_encode_kernel(const int64_t* __restrict__ offsets,
                                  const uint8_t* __restrict__ bytes,
                                  const int32_t* __restrict__ byte2id,
                                  const uint64_t* __restrict__ tkeys,
                                  const uint64_t* __restrict__ tvals,
                                  int tsize_mask, int64_t n_rows,
                                  int32_t* __restrict__ out_ids,
                                  int32_t* __restrict__ out_counts) {
  __shared__ int32_t ids[BPE_MAX_LEN];
  const int lane = threadIdx.x;                 // blockDim.x == WAVE
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const int64_t lo = offsets[row];
    const int64_t hi = offsets[row + 1];
    int L = (int)(hi - lo);
    if (L > BPE_MAX_LEN) {
      if (lane == 0) out_counts[row] = -1;      // host fallback
      __syncthreads();
      continue;
    }
    for (int i = lane; i < L; i += WAVE)
      ids[i] = byte2id[bytes[lo + i]];
    __syncthreads();

    while (L > 1) {
      uint32_t best_rank = RANK_INF;
      int best_pos = -1;
      for (int i = lane; i < L - 1; i += WAVE) {
        uint64_t key = ((uint64_t)(uint32_t)ids[i] << 32) |
                       (uint32_t)ids[i + 1];
        uint32_t r, nid;
        pair_lookup(tkeys, tvals, tsize_mask, key, &r, &nid);
        // tie-break on the LEFTMOST position for a given rank
        if (r < best_rank || (r == best_rank && i < best_pos)) {
          best_rank = r;
          best_pos = i;
        }
      }
      // wave min-reduction on (rank, pos): pack rank into high bits so a
      // single u64 min gives leftmost-lowest-rank
      uint64_t packed = ((uint64_t)best_rank << 32) |
                        (uint32_t)(best_pos < 0 ? 0x7FFFFFFF : best_pos);
      for (int off = WAVE / 2; off; off >>= 1) {
        uint64_t other = __shfl_down(packed, off, WAVE);
        if (other < packed) packed = other;
      }
      packed = __shfl(packed, 0, WAVE);
      uint32_t rank = (uint32_t)(packed >> 32);
      if (rank == RANK_INF) break;
      int pos = (int)(uint32_t)(packed & 0x7FFFFFFF);
      if (lane == 0) {
        uint64_t key = ((uint64_t)(uint32_t)ids[pos] << 32) |
                       (uint32_t)ids[pos + 1];
        uint32_t r, nid;
        pair_lookup(tkeys, tvals, tsize_mask, key, &r, &nid);
        ids[pos] = (int32_t)nid;
      }
      __syncthreads();
      // shift the tail left by one (two-phase to avoid LDS RAW hazards)
      for (int base = pos + 1; base < L - 1; base += WAVE) {
        int i = base + lane;
        int32_t v = 0;
        bool act = i < L - 1;
        if (act) v = ids[i + 1];
        __syncthreads();
        if (act) ids[i] = v;
        __syncthreads();
      }
      --L;
    }

    for (int i = lane; i < L; i += WAVE) out_ids[lo + i] = ids[i];
    if (lane == 0) out_counts[row] = L;
    __syncthreads();
  }
}

}  // namespace

std::vector<Tensor> bpe_encode(Tensor offsets, Tensor bytes, Tensor byte2id,
                               Tensor table_keys, Tensor table_vals) {
  int64_t n = offsets.numel() - 1;
  auto dev = bytes.device();
  auto out_ids = torch::zeros({std::max<int64_t>(bytes.numel(), 1)},
                              torch::dtype(torch::kInt32).device(dev));
  auto out_counts = torch::zeros({std::max<int64_t>(n, 1)},
                                 torch::dtype(torch::kInt32).device(dev));
  if (n == 0) return {out_ids, out_counts};
  int tsize_mask = (int)table_keys.numel() - 1;
  int grid = (int)std::min<int64_t>(n, 4096);
  hipLaunchKernelGGL(bpe_encode_kernel, dim3(grid), dim3(WAVE), 0,
                     at::hip::getCurrentHIPStream().stream(),
                     offsets.data_ptr<int64_t>(),
                     (const uint8_t*)bytes.data_ptr(),
                     byte2id.data_ptr<int32_t>(),
                     (const uint64_t*)table_keys.data_ptr<int64_t>(),
                     (const uint64_t*)table_vals.data_ptr<int64_t>(),
                     tsize_mask, n, out_ids.data_ptr<int32_t>(),
                     out_counts.data_ptr<int32_t>());
  return {out_ids, out_counts};
}
