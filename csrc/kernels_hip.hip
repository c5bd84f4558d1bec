#include "hip/hip_runtime.h"
// Core row-wise HIP kernels for gfx950: row hashing, stream compaction,
// string gather, hash groupby (LDS partial agg), bucket-chain hash join,
// hash partitioning.  MI355X-native design notes:
//  * one thread per row, grid-stride, grids capped at 8 blocks/CU
//  * hash tables are power-of-2 bucket arrays in HBM; inserts use one
//    device-scope atomic each (no locks) — per-XCD L2 non-coherence is safe
//    because all cross-workgroup communication is via atomics
//  * low-cardinality aggregations stage per-block accumulators in LDS
//    (160 KiB/CU) and emit one global atomic per block per group
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"
#include "row_ops.h"
#include "api.h"

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
// descriptor packing
// ---------------------------------------------------------------------------

Tensor pack_descs(const std::vector<int64_t>& tags,
                  const std::vector<Tensor>& datas,
                  const std::vector<OptTensor>& offsets,
                  const std::vector<OptTensor>& validities) {
  int n = (int)tags.size();
  auto host = torch::empty({n * 4}, torch::dtype(torch::kInt64));
  int64_t* h = host.data_ptr<int64_t>();
  for (int i = 0; i < n; ++i) {
    h[i * 4 + 0] = (int64_t)datas[i].data_ptr();
    h[i * 4 + 1] = offsets[i].has_value()
                       ? (int64_t)offsets[i]->data_ptr<int64_t>()
                       : 0;
    h[i * 4 + 2] = validities[i].has_value()
                       ? (int64_t)validities[i]->data_ptr<bool>()
                       : 0;
    h[i * 4 + 3] = tags[i];
  }
  return host.to(datas[0].device(), /*non_blocking=*/false);
}

// ---------------------------------------------------------------------------
// row hashing
// ---------------------------------------------------------------------------

__global__ void hash_rows_kernel(const ColDesc* cols, int ncols, int64_t n,
                                 uint64_t seed, uint64_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    out[i] = row_hash(cols, ncols, i, seed);
  }
}

Tensor hash_rows(const std::vector<int64_t>& tags,
                 const std::vector<Tensor>& datas,
                 const std::vector<OptTensor>& offsets,
                 const std::vector<OptTensor>& validities, int64_t n,
                 int64_t seed) {
  auto descs = pack_descs(tags, datas, offsets, validities);
  auto out = torch::empty({n}, torch::dtype(torch::kInt64)
                                   .device(datas[0].device()));
  if (n == 0) return out;
  int block = 256;
  hipLaunchKernelGGL(hash_rows_kernel, dim3(grid_1d(n, block)), dim3(block),
                     0, cur_stream(), (const ColDesc*)descs.data_ptr(),
                     (int)tags.size(), n, (uint64_t)seed,
                     (uint64_t*)out.data_ptr<int64_t>());
  return out;
}

// ---------------------------------------------------------------------------
// stream compaction: bool mask -> selected indices
// (ballot + wave prefix within block, block offsets via device cumsum)
// ---------------------------------------------------------------------------

__global__ void block_count_kernel(const bool* mask, int64_t n,
                                   int64_t chunk, int32_t* counts) {
  int64_t begin = (int64_t)blockIdx.x * chunk;
  int64_t end = min(begin + chunk, n);
  int32_t local = 0;
  for (int64_t i = begin + threadIdx.x; i < end; i += blockDim.x)
    local += mask[i] ? 1 : 0;
  __shared__ int32_t acc;
  if (threadIdx.x == 0) acc = 0;
  __syncthreads();
  // wave reduce then one atomic per wave
  for (int off = WAVE / 2; off; off >>= 1)
    local += __shfl_down(local, off, WAVE);
  if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(&acc, local);
  __syncthreads();
  if (threadIdx.x == 0) counts[blockIdx.x] = acc;
}

__global__ void compact_kernel(const bool* mask, int64_t n, int64_t chunk,
                               const int32_t* block_offsets, int64_t* out) {
  int64_t begin = (int64_t)blockIdx.x * chunk;
  int64_t end = min(begin + chunk, n);
  __shared__ int64_t base;
  __shared__ int32_t wave_counts[256 / WAVE];
  if (threadIdx.x == 0) base = block_offsets[blockIdx.x];
  __syncthreads();
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nwaves = blockDim.x / WAVE;
  for (int64_t tile = begin; tile < end; tile += blockDim.x) {
    int64_t i = tile + threadIdx.x;
    bool sel = i < end && mask[i];
    uint64_t ballot = __ballot(sel);
    int rank = __popcll(ballot & ((1ull << lane) - 1));
    int wtotal = __popcll(ballot);
    if (lane == 0) wave_counts[wid] = wtotal;
    __syncthreads();
    int woff = 0;
    for (int w = 0; w < wid; ++w) woff += wave_counts[w];
    if (sel) out[base + woff + rank] = i;
    __syncthreads();
    if (threadIdx.x == 0) {
      int t = 0;
      for (int w = 0; w < nwaves; ++w) t += wave_counts[w];
      base += t;
    }
    __syncthreads();
  }
}

Tensor compact_indices(Tensor mask) {
  TORCH_CHECK(mask.is_cuda() && mask.dtype() == torch::kBool);
  mask = mask.contiguous();
  int64_t n = mask.numel();
  auto dev = mask.device();
  if (n == 0) return torch::empty({0}, torch::dtype(torch::kInt64).device(dev));
  int block = 256;
  int64_t chunk = (n + kMaxBlocks - 1) / kMaxBlocks;
  if (chunk < block) chunk = block;
  int nblocks = (int)((n + chunk - 1) / chunk);
  auto counts = torch::empty({nblocks},
                             torch::dtype(torch::kInt32).device(dev));
  hipLaunchKernelGGL(block_count_kernel, dim3(nblocks), dim3(block), 0,
                     cur_stream(), mask.data_ptr<bool>(), n, chunk,
                     counts.data_ptr<int32_t>());
  auto offs = torch::cumsum(counts, 0, torch::kInt32) - counts;
  auto total_t = counts.sum();
  int64_t total = total_t.item<int64_t>();
  auto out = torch::empty({total}, torch::dtype(torch::kInt64).device(dev));
  if (total == 0) return out;
  hipLaunchKernelGGL(compact_kernel, dim3(nblocks), dim3(block), 0,
                     cur_stream(), mask.data_ptr<bool>(), n, chunk,
                     offs.contiguous().data_ptr<int32_t>(),
                     out.data_ptr<int64_t>());
  return out;
}

// ---------------------------------------------------------------------------
// string gather: one thread per OUTPUT BYTE (coalesced writes; row lookup via
// binary search over output offsets, L2/LDS-friendly)
// ---------------------------------------------------------------------------

DEV_INLINE int64_t upper_bound_row(const int64_t* offs, int64_t nrows,
                                   int64_t j) {
  int64_t lo = 0, hi = nrows;  // offs has nrows+1 entries; find row of byte j
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (offs[mid + 1] <= j)
      lo = mid + 1;
    else
      hi = mid;
  }
  return lo;
}

__global__ void gather_bytes_kernel(const int64_t* src_off,
                                    const uint8_t* src_bytes,
                                    const int64_t* idx,
                                    const int64_t* out_off, int64_t nrows,
                                    int64_t total, uint8_t* out) {
  // One binary search per 256-byte block (LDS-shared), then threads walk
  // offsets linearly — offsets are monotone so the walk is short and the
  // reads coalesce within the block.
  __shared__ int64_t block_row;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j0 = (int64_t)blockIdx.x * blockDim.x; j0 < total;
       j0 += stride) {
    if (threadIdx.x == 0) block_row = upper_bound_row(out_off, nrows, j0);
    __syncthreads();
    int64_t j = j0 + threadIdx.x;
    if (j < total) {
      int64_t row = block_row;
      while (out_off[row + 1] <= j) ++row;
      int64_t within = j - out_off[row];
      int64_t src_row = idx[row];
      out[j] = src_bytes[src_off[src_row] + within];
    }
    __syncthreads();
  }
}

std::vector<Tensor> take_string(Tensor offsets, Tensor bytes, Tensor idx) {
  auto dev = offsets.device();
  int64_t m = idx.numel();
  auto lens = offsets.slice(0, 1) - offsets.slice(0, 0, offsets.numel() - 1);
  auto sel_lens = lens.index_select(0, idx);
  auto out_off = torch::zeros({m + 1}, torch::dtype(torch::kInt64).device(dev));
  if (m > 0)
    out_off.slice(0, 1).copy_(torch::cumsum(sel_lens, 0));
  int64_t total = m > 0 ? out_off[m].item<int64_t>() : 0;
  auto out = torch::empty({total}, torch::dtype(torch::kUInt8).device(dev));
  if (total > 0) {
    int block = 256;
    hipLaunchKernelGGL(gather_bytes_kernel, dim3(grid_1d(total, block)),
                       dim3(block), 0, cur_stream(),
                       offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
                       idx.data_ptr<int64_t>(), out_off.data_ptr<int64_t>(),
                       m, total, out.data_ptr<uint8_t>());
  }
  return {out_off, out};
}

__global__ void merge_strings_kernel(const bool* m, const int64_t* t_off,
                                     const uint8_t* t_bytes,
                                     const int64_t* f_off,
                                     const uint8_t* f_bytes,
                                     const int64_t* out_off, int64_t nrows,
                                     int64_t total, uint8_t* out) {
  __shared__ int64_t block_row;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j0 = (int64_t)blockIdx.x * blockDim.x; j0 < total;
       j0 += stride) {
    if (threadIdx.x == 0) block_row = upper_bound_row(out_off, nrows, j0);
    __syncthreads();
    int64_t j = j0 + threadIdx.x;
    if (j < total) {
      int64_t row = block_row;
      while (out_off[row + 1] <= j) ++row;
      int64_t within = j - out_off[row];
      out[j] = m[row] ? t_bytes[t_off[row] + within]
                      : f_bytes[f_off[row] + within];
    }
    __syncthreads();
  }
}

Tensor merge_strings(Tensor mask, Tensor t_off, Tensor t_bytes, Tensor f_off,
                     Tensor f_bytes, Tensor out_off) {
  int64_t nrows = mask.numel();
  int64_t total = nrows > 0 ? out_off[nrows].item<int64_t>() : 0;
  auto out = torch::empty({total},
                          torch::dtype(torch::kUInt8).device(mask.device()));
  if (total > 0) {
    int block = 256;
    hipLaunchKernelGGL(merge_strings_kernel, dim3(grid_1d(total, block)),
                       dim3(block), 0, cur_stream(), mask.data_ptr<bool>(),
                       t_off.data_ptr<int64_t>(), t_bytes.data_ptr<uint8_t>(),
                       f_off.data_ptr<int64_t>(), f_bytes.data_ptr<uint8_t>(),
                       out_off.data_ptr<int64_t>(), nrows, total,
                       out.data_ptr<uint8_t>());
  }
  return out;
}

// ---------------------------------------------------------------------------
// hash groupby: open-addressing insert keyed by row hash with full key verify
// (ref semantics: daft-recordbatch probe_table.rs; GPU design: linear-probe
// table of owner row indices, dense ids assigned by atomic counter)
// ---------------------------------------------------------------------------

__global__ void groupby_insert_kernel(const uint64_t* hashes,
                                      const ColDesc* cols, int ncols,
                                      int64_t n, int64_t* table,
                                      uint64_t mask, int64_t* row_slot) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t h = hashes[i];
    uint64_t slot = h & mask;
    while (true) {
      // Plain read first: slots transition -1 -> owner exactly once, so a
      // non-(-1) read is final; a stale -1 is corrected by the device-scope
      // CAS below.  This avoids serializing 10^8 CAS ops on a handful of
      // slots for low-cardinality groupbys (the Q1 shape).
      long long prev = table[slot];
      if (prev == -1ll) {
        prev = atomicCAS((unsigned long long*)&table[slot],
                         (unsigned long long)(-1ll),
                         (unsigned long long)i);
        if (prev == -1ll) {  // claimed: i is the representative
          row_slot[i] = (int64_t)slot;
          break;
        }
      }
      if (hashes[prev] == h &&
          row_eq(cols, cols, ncols, prev, i, /*null_eq=*/true)) {
        row_slot[i] = (int64_t)slot;
        break;
      }
      slot = (slot + 1) & mask;
    }
  }
}

__global__ void groupby_assign_ids_kernel(const int64_t* table,
                                          const int64_t* row_slot, int64_t n,
                                          int32_t* slot_gid, int64_t* reps,
                                          int32_t* counter) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t slot = row_slot[i];
    if (table[slot] == i) {
      int32_t gid = atomicAdd(counter, 1);
      slot_gid[slot] = gid;
      reps[gid] = i;
    }
  }
}

__global__ void groupby_gather_ids_kernel(const int64_t* row_slot,
                                          const int32_t* slot_gid, int64_t n,
                                          int64_t* gids) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    gids[i] = (int64_t)slot_gid[row_slot[i]];
}

static int64_t table_capacity(int64_t n) {
  int64_t cap = 64;
  while (cap < 2 * n) cap <<= 1;
  return cap;
}

std::vector<Tensor> groupby(Tensor hashes, const std::vector<int64_t>& tags,
                            const std::vector<Tensor>& datas,
                            const std::vector<OptTensor>& offsets,
                            const std::vector<OptTensor>& validities) {
  auto dev = hashes.device();
  int64_t n = hashes.numel();
  auto opts64 = torch::dtype(torch::kInt64).device(dev);
  if (n == 0) {
    return {torch::empty({0}, opts64), torch::empty({0}, opts64)};
  }
  auto descs = pack_descs(tags, datas, offsets, validities);
  int64_t cap = table_capacity(n);
  auto table = torch::full({cap}, -1, opts64);
  auto row_slot = torch::empty({n}, opts64);
  int block = 256;
  hipLaunchKernelGGL(groupby_insert_kernel, dim3(grid_1d(n, block)),
                     dim3(block), 0, cur_stream(),
                     (const uint64_t*)hashes.data_ptr<int64_t>(),
                     (const ColDesc*)descs.data_ptr(), (int)tags.size(), n,
                     table.data_ptr<int64_t>(), (uint64_t)(cap - 1),
                     row_slot.data_ptr<int64_t>());
  auto slot_gid = torch::empty({cap}, torch::dtype(torch::kInt32).device(dev));
  auto reps_full = torch::empty({n}, opts64);
  auto counter = torch::zeros({1}, torch::dtype(torch::kInt32).device(dev));
  hipLaunchKernelGGL(groupby_assign_ids_kernel, dim3(grid_1d(n, block)),
                     dim3(block), 0, cur_stream(), table.data_ptr<int64_t>(),
                     row_slot.data_ptr<int64_t>(), n,
                     slot_gid.data_ptr<int32_t>(),
                     reps_full.data_ptr<int64_t>(),
                     counter.data_ptr<int32_t>());
  auto gids = torch::empty({n}, opts64);
  hipLaunchKernelGGL(groupby_gather_ids_kernel, dim3(grid_1d(n, block)),
                     dim3(block), 0, cur_stream(),
                     row_slot.data_ptr<int64_t>(),
                     slot_gid.data_ptr<int32_t>(), n,
                     gids.data_ptr<int64_t>());
  int64_t num_groups = counter.item<int32_t>();
  return {gids, reps_full.slice(0, 0, num_groups)};
}

// ---------------------------------------------------------------------------
// grouped aggregation: LDS partial accumulators when num_groups is small,
// global atomics otherwise (low contention at high cardinality)
// ---------------------------------------------------------------------------

DEV_INLINE uint64_t f64_order_bits(double v) {
  uint64_t b;
  __builtin_memcpy(&b, &v, 8);
  return (b & 0x8000000000000000ull) ? ~b : (b ^ 0x8000000000000000ull);
}
DEV_INLINE double f64_from_order_bits(uint64_t b) {
  b = (b & 0x8000000000000000ull) ? (b ^ 0x8000000000000000ull) : ~b;
  double v;
  __builtin_memcpy(&v, &b, 8);
  return v;
}


template <typename T>
DEV_INLINE void atomic_add_val(T* p, T v);
template <>
DEV_INLINE void atomic_add_val<double>(double* p, double v) {
  atomicAdd(p, v);
}
template <>
DEV_INLINE void atomic_add_val<int64_t>(int64_t* p, int64_t v) {
  atomicAdd((unsigned long long*)p, (unsigned long long)v);
}

// OP: 0 sum, 1 min, 2 max.  FP: double or int64 values.
template <typename T, int OP>
__global__ void grouped_agg_global_kernel(const int64_t* gids, const T* vals,
                                          const bool* valid, int64_t n,
                                          T* out, int64_t* cnt) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (valid && !valid[i]) continue;
    int64_t g = gids[i];
    atomicAdd((unsigned long long*)&cnt[g], 1ull);
    if constexpr (OP == 0) {
      atomic_add_val(&out[g], vals[i]);
    } else if constexpr (std::is_same<T, double>::value) {
      // ordered-bits min/max: out holds order-encoded u64
      uint64_t ob = f64_order_bits((double)vals[i]);
      if constexpr (OP == 1)
        atomicMin((unsigned long long*)&out[g], (unsigned long long)ob);
      else
        atomicMax((unsigned long long*)&out[g], (unsigned long long)ob);
    } else {
      if constexpr (OP == 1)
        atomicMin((long long*)&out[g], (long long)vals[i]);
      else
        atomicMax((long long*)&out[g], (long long)vals[i]);
    }
  }
}

template <typename T, int OP>
__global__ void grouped_agg_lds_kernel(const int64_t* gids, const T* vals,
                                       const bool* valid, int64_t n,
                                       int64_t num_groups, T* out,
                                       int64_t* cnt) {
  extern __shared__ char smem[];
  T* lacc = (T*)smem;
  uint32_t* lcnt = (uint32_t*)(smem + num_groups * sizeof(T));
  for (int64_t g = threadIdx.x; g < num_groups; g += blockDim.x) {
    if constexpr (OP == 0)
      lacc[g] = (T)0;
    else if constexpr (std::is_same<T, double>::value)
      ((uint64_t*)lacc)[g] = OP == 1 ? ~0ull : 0ull;  // order-bits identity
    else
      lacc[g] = OP == 1 ? (T)INT64_MAX : (T)INT64_MIN;
    lcnt[g] = 0;
  }
  __syncthreads();
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (valid && !valid[i]) continue;
    int64_t g = gids[i];
    atomicAdd(&lcnt[g], 1u);
    if constexpr (OP == 0) {
      atomic_add_val(&lacc[g], vals[i]);
    } else if constexpr (std::is_same<T, double>::value) {
      uint64_t ob = f64_order_bits((double)vals[i]);
      if constexpr (OP == 1)
        atomicMin((unsigned long long*)&lacc[g], (unsigned long long)ob);
      else
        atomicMax((unsigned long long*)&lacc[g], (unsigned long long)ob);
    } else {
      if constexpr (OP == 1)
        atomicMin((long long*)&lacc[g], (long long)vals[i]);
      else
        atomicMax((long long*)&lacc[g], (long long)vals[i]);
    }
  }
  __syncthreads();
  for (int64_t g = threadIdx.x; g < num_groups; g += blockDim.x) {
    if (lcnt[g] == 0) continue;
    atomicAdd((unsigned long long*)&cnt[g], (unsigned long long)lcnt[g]);
    if constexpr (OP == 0) {
      atomic_add_val(&out[g], lacc[g]);
    } else if constexpr (std::is_same<T, double>::value) {
      if constexpr (OP == 1)
        atomicMin((unsigned long long*)&out[g], ((uint64_t*)lacc)[g]);
      else
        atomicMax((unsigned long long*)&out[g], ((uint64_t*)lacc)[g]);
    } else {
      if constexpr (OP == 1)
        atomicMin((long long*)&out[g], (long long)lacc[g]);
      else
        atomicMax((long long*)&out[g], (long long)lacc[g]);
    }
  }
}

__global__ void decode_f64_order_kernel(uint64_t* buf, const int64_t* cnt,
                                        int64_t n, double fill) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    double v = cnt[i] > 0 ? f64_from_order_bits(buf[i]) : fill;
    __builtin_memcpy(&buf[i], &v, 8);
  }
}

template <typename T, int OP>
static void launch_agg(const int64_t* gids, const T* vals, const bool* valid,
                       int64_t n, int64_t num_groups, T* out, int64_t* cnt) {
  int block = 256;
  // LDS budget: keep under 48 KiB/block for occupancy (160 KiB/CU on CDNA4)
  int64_t lds_bytes = num_groups * (sizeof(T) + sizeof(uint32_t));
  if (num_groups <= 4096 && lds_bytes <= 48 * 1024) {
    hipLaunchKernelGGL((grouped_agg_lds_kernel<T, OP>),
                       dim3(grid_1d(n, block, 8)), dim3(block), lds_bytes,
                       cur_stream(), gids, vals, valid, n, num_groups, out,
                       cnt);
  } else {
    hipLaunchKernelGGL((grouped_agg_global_kernel<T, OP>),
                       dim3(grid_1d(n, block)), dim3(block), 0, cur_stream(),
                       gids, vals, valid, n, out, cnt);
  }
}

std::vector<Tensor> grouped_agg(Tensor group_ids, int64_t num_groups,
                                Tensor values, Tensor valid,
                                const std::string& op) {
  auto dev = group_ids.device();
  int64_t n = group_ids.numel();
  bool is_f64 = values.dtype() == torch::kFloat64;
  TORCH_CHECK(is_f64 || values.dtype() == torch::kInt64,
              "grouped_agg expects f64/i64 working dtype");
  auto cnt = torch::zeros({num_groups}, torch::dtype(torch::kInt64).device(dev));
  Tensor out;
  int opi = op == "sum" ? 0 : op == "min" ? 1 : 2;
  if (opi == 0) {
    out = torch::zeros({num_groups}, values.options());
  } else if (is_f64) {
    // order-encoded identity
    out = torch::full({num_groups}, (int64_t)(opi == 1 ? -1 : 0),
                      torch::dtype(torch::kInt64).device(dev));
  } else {
    out = torch::full({num_groups},
                      opi == 1 ? INT64_MAX : INT64_MIN,
                      torch::dtype(torch::kInt64).device(dev));
  }
  if (n > 0) {
    const bool* vp = (valid.defined() && valid.numel() > 0)
                         ? valid.data_ptr<bool>() : nullptr;
    const int64_t* g = group_ids.data_ptr<int64_t>();
    if (is_f64) {
      const double* v = values.data_ptr<double>();
      double* o = opi == 0 ? out.data_ptr<double>()
                           : (double*)out.data_ptr<int64_t>();
      if (opi == 0)
        launch_agg<double, 0>(g, v, vp, n, num_groups, o,
                              cnt.data_ptr<int64_t>());
      else if (opi == 1)
        launch_agg<double, 1>(g, v, vp, n, num_groups, o,
                              cnt.data_ptr<int64_t>());
      else
        launch_agg<double, 2>(g, v, vp, n, num_groups, o,
                              cnt.data_ptr<int64_t>());
    } else {
      const int64_t* v = values.data_ptr<int64_t>();
      int64_t* o = out.data_ptr<int64_t>();
      if (opi == 0)
        launch_agg<int64_t, 0>(g, v, vp, n, num_groups, o,
                               cnt.data_ptr<int64_t>());
      else if (opi == 1)
        launch_agg<int64_t, 1>(g, v, vp, n, num_groups, o,
                               cnt.data_ptr<int64_t>());
      else
        launch_agg<int64_t, 2>(g, v, vp, n, num_groups, o,
                               cnt.data_ptr<int64_t>());
    }
  }
  if (is_f64 && opi != 0 && num_groups > 0) {
    int block = 256;
    int grid = (int)((num_groups + block - 1) / block);
    hipLaunchKernelGGL(decode_f64_order_kernel, dim3(grid), dim3(block), 0,
                       cur_stream(), (uint64_t*)out.data_ptr<int64_t>(),
                       cnt.data_ptr<int64_t>(), num_groups, 0.0);
    out = out.view(torch::kFloat64);
  }
  return {out, cnt};
}

__global__ void grouped_count_lds_kernel(const int64_t* gids,
                                         const bool* valid, int64_t n,
                                         int64_t num_groups, int64_t* cnt) {
  extern __shared__ uint32_t lcnt[];
  for (int64_t g = threadIdx.x; g < num_groups; g += blockDim.x) lcnt[g] = 0;
  __syncthreads();
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (valid && !valid[i]) continue;
    atomicAdd(&lcnt[gids[i]], 1u);
  }
  __syncthreads();
  for (int64_t g = threadIdx.x; g < num_groups; g += blockDim.x)
    if (lcnt[g])
      atomicAdd((unsigned long long*)&cnt[g], (unsigned long long)lcnt[g]);
}

__global__ void grouped_count_global_kernel(const int64_t* gids,
                                            const bool* valid, int64_t n,
                                            int64_t* cnt) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (valid && !valid[i]) continue;
    atomicAdd((unsigned long long*)&cnt[gids[i]], 1ull);
  }
}

Tensor grouped_count(Tensor group_ids, int64_t num_groups, Tensor valid) {
  auto dev = group_ids.device();
  int64_t n = group_ids.numel();
  auto cnt = torch::zeros({num_groups},
                          torch::dtype(torch::kInt64).device(dev));
  if (n == 0) return cnt;
  const bool* vp = (valid.defined() && valid.numel() > 0)
                       ? valid.data_ptr<bool>() : nullptr;
  int block = 256;
  int64_t lds = num_groups * sizeof(uint32_t);
  if (num_groups <= 8192 && lds <= 48 * 1024) {
    hipLaunchKernelGGL(grouped_count_lds_kernel,
                       dim3(grid_1d(n, block, 8)), dim3(block), lds,
                       cur_stream(), group_ids.data_ptr<int64_t>(), vp, n,
                       num_groups, cnt.data_ptr<int64_t>());
  } else {
    hipLaunchKernelGGL(grouped_count_global_kernel,
                       dim3(grid_1d(n, block)), dim3(block), 0, cur_stream(),
                       group_ids.data_ptr<int64_t>(), vp, n,
                       cnt.data_ptr<int64_t>());
  }
  return cnt;
}

// one-pass count-distinct: insert (group, value) pairs into an
// open-addressing table; the CAS winner (first sighting of a distinct pair)
// increments its group's counter.  No dense ids / rep gathers needed.
__global__ void count_distinct_kernel(const uint64_t* hashes,
                                      const ColDesc* cols, int ncols,
                                      const int64_t* gids, int64_t n,
                                      int64_t* table, uint64_t mask,
                                      int64_t* cnt) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (row_has_null(cols, ncols, i)) continue;  // nulls don't count
    uint64_t h = hashes[i];
    uint64_t slot = h & mask;
    while (true) {
      long long prev = table[slot];
      if (prev == -1ll) {
        prev = atomicCAS((unsigned long long*)&table[slot],
                         (unsigned long long)(-1ll), (unsigned long long)i);
        if (prev == -1ll) {
          atomicAdd((unsigned long long*)&cnt[gids[i]], 1ull);
          break;
        }
      }
      if (hashes[prev] == h && row_eq(cols, cols, ncols, prev, i, true))
        break;
      slot = (slot + 1) & mask;
    }
  }
}

Tensor count_distinct_pairs(Tensor hashes, const std::vector<int64_t>& tags,
                            const std::vector<Tensor>& datas,
                            const std::vector<OptTensor>& offsets,
                            const std::vector<OptTensor>& validities,
                            Tensor gids, int64_t num_groups) {
  auto dev = hashes.device();
  int64_t n = hashes.numel();
  auto cnt = torch::zeros({num_groups},
                          torch::dtype(torch::kInt64).device(dev));
  if (n == 0) return cnt;
  auto descs = pack_descs(tags, datas, offsets, validities);
  int64_t cap = table_capacity(n);
  auto table = torch::full({cap}, -1,
                           torch::dtype(torch::kInt64).device(dev));
  int block = 256;
  hipLaunchKernelGGL(count_distinct_kernel, dim3(grid_1d(n, block)),
                     dim3(block), 0, cur_stream(),
                     (const uint64_t*)hashes.data_ptr<int64_t>(),
                     (const ColDesc*)descs.data_ptr(), (int)tags.size(),
                     gids.data_ptr<int64_t>(), n,
                     table.data_ptr<int64_t>(), (uint64_t)(cap - 1),
                     cnt.data_ptr<int64_t>());
  return cnt;
}

// dense-range groupby support: first-occurrence index per packed key
// (LDS-staged atomicMin for small ranges — global atomics on a handful of
// addresses would serialize)
__global__ void dense_first_lds_kernel(const int64_t* packed, int64_t n,
                                       int64_t rng, int64_t* first) {
  extern __shared__ uint32_t lmin[];
  for (int64_t g = threadIdx.x; g < rng; g += blockDim.x)
    lmin[g] = 0xFFFFFFFFu;
  __syncthreads();
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    atomicMin(&lmin[packed[i]], (uint32_t)i);
  __syncthreads();
  for (int64_t g = threadIdx.x; g < rng; g += blockDim.x)
    if (lmin[g] != 0xFFFFFFFFu)
      atomicMin((unsigned long long*)&first[g],
                (unsigned long long)lmin[g]);
}

__global__ void dense_first_global_kernel(const int64_t* packed, int64_t n,
                                          int64_t* first) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    atomicMin((unsigned long long*)&first[packed[i]],
              (unsigned long long)i);
}

Tensor dense_first_index(Tensor packed, int64_t rng, int64_t sentinel) {
  auto dev = packed.device();
  int64_t n = packed.numel();
  auto first = torch::full({rng}, sentinel,
                           torch::dtype(torch::kInt64).device(dev));
  if (n == 0) return first;
  TORCH_CHECK(n < (int64_t)0xFFFFFFFF, "dense_first_index: n must fit u32");
  int block = 256;
  int64_t lds = rng * sizeof(uint32_t);
  if (rng <= 8192 && lds <= 48 * 1024) {
    hipLaunchKernelGGL(dense_first_lds_kernel, dim3(grid_1d(n, block, 8)),
                       dim3(block), lds, cur_stream(),
                       packed.data_ptr<int64_t>(), n, rng,
                       first.data_ptr<int64_t>());
  } else {
    hipLaunchKernelGGL(dense_first_global_kernel,
                       dim3(grid_1d(n, block)), dim3(block), 0, cur_stream(),
                       packed.data_ptr<int64_t>(), n,
                       first.data_ptr<int64_t>());
  }
  return first;
}

// ---------------------------------------------------------------------------
// hash join: bucket-chain build + verified probe
// ---------------------------------------------------------------------------

__global__ void join_build_kernel(const uint64_t* hashes, int64_t n,
                                  int64_t* heads, int64_t* next,
                                  uint64_t mask) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t b = hashes[i] & mask;
    long long prev = atomicExch((unsigned long long*)&heads[b],
                                (unsigned long long)i);
    next[i] = prev;
  }
}

std::vector<Tensor> join_build(Tensor hashes) {
  auto dev = hashes.device();
  int64_t n = hashes.numel();
  auto opts64 = torch::dtype(torch::kInt64).device(dev);
  int64_t cap = table_capacity(std::max<int64_t>(n, 1));
  auto heads = torch::full({cap}, -1, opts64);
  auto next = torch::full({std::max<int64_t>(n, 1)}, -1, opts64);
  if (n > 0) {
    int block = 256;
    hipLaunchKernelGGL(join_build_kernel, dim3(grid_1d(n, block)),
                       dim3(block), 0, cur_stream(),
                       (const uint64_t*)hashes.data_ptr<int64_t>(), n,
                       heads.data_ptr<int64_t>(), next.data_ptr<int64_t>(),
                       (uint64_t)(cap - 1));
  }
  return {heads, next};
}

// mode: 0=inner, 1=left, 2=semi, 3=anti
__global__ void join_count_kernel(const int64_t* heads, const int64_t* next,
                                  const uint64_t* ph, const uint64_t* bh,
                                  const ColDesc* pc, const ColDesc* bc,
                                  int ncols, int64_t np, uint64_t mask,
                                  int mode, int32_t* counts) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < np;
       i += stride) {
    uint64_t h = ph[i];
    int64_t m = 0;
    if (!row_has_null(pc, ncols, i)) {
      int64_t j = heads[h & mask];
      while (j != -1) {
        if (bh[j] == h && row_eq(pc, bc, ncols, i, j, false)) {
          ++m;
          if (mode >= 2) break;  // semi/anti: existence only
        }
        j = next[j];
      }
    }
    int32_t c;
    switch (mode) {
      case 0: c = (int32_t)m; break;
      case 1: c = (int32_t)(m ? m : 1); break;
      case 2: c = m ? 1 : 0; break;
      default: c = m ? 0 : 1; break;
    }
    counts[i] = c;
  }
}

__global__ void join_fill_kernel(const int64_t* heads, const int64_t* next,
                                 const uint64_t* ph, const uint64_t* bh,
                                 const ColDesc* pc, const ColDesc* bc,
                                 int ncols, int64_t np, uint64_t mask,
                                 int mode, const int32_t* offs, int64_t* lidx,
                                 int64_t* ridx, bool* matched) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < np;
       i += stride) {
    uint64_t h = ph[i];
    int64_t pos = offs[i];
    int64_t m = 0;
    if (!row_has_null(pc, ncols, i)) {
      int64_t j = heads[h & mask];
      while (j != -1) {
        if (bh[j] == h && row_eq(pc, bc, ncols, i, j, false)) {
          ++m;
          if (mode >= 2) break;
          lidx[pos] = i;
          ridx[pos] = j;
          matched[j] = true;
          ++pos;
        }
        j = next[j];
      }
    }
    if (mode == 1 && m == 0) {
      lidx[pos] = i;
      ridx[pos] = -1;
    } else if (mode == 2 && m) {
      lidx[pos] = i;
    } else if (mode == 3 && !m) {
      lidx[pos] = i;
    }
  }
}

std::vector<Tensor> join_probe(
    Tensor table, Tensor next, Tensor probe_hashes, Tensor build_hashes,
    const std::vector<int64_t>& tags_l, const std::vector<Tensor>& data_l,
    const std::vector<OptTensor>& off_l, const std::vector<OptTensor>& val_l,
    const std::vector<int64_t>& tags_r, const std::vector<Tensor>& data_r,
    const std::vector<OptTensor>& off_r, const std::vector<OptTensor>& val_r,
    int64_t mode) {
  auto dev = probe_hashes.device();
  int64_t np = probe_hashes.numel();
  int64_t nb = build_hashes.numel();
  auto opts64 = torch::dtype(torch::kInt64).device(dev);
  auto pdescs = pack_descs(tags_l, data_l, off_l, val_l);
  auto bdescs = pack_descs(tags_r, data_r, off_r, val_r);
  uint64_t mask = (uint64_t)table.numel() - 1;
  auto counts = torch::zeros({std::max<int64_t>(np, 1)},
                             torch::dtype(torch::kInt32).device(dev));
  int block = 256;
  if (np > 0) {
    hipLaunchKernelGGL(join_count_kernel, dim3(grid_1d(np, block)),
                       dim3(block), 0, cur_stream(),
                       table.data_ptr<int64_t>(), next.data_ptr<int64_t>(),
                       (const uint64_t*)probe_hashes.data_ptr<int64_t>(),
                       (const uint64_t*)build_hashes.data_ptr<int64_t>(),
                       (const ColDesc*)pdescs.data_ptr(),
                       (const ColDesc*)bdescs.data_ptr(), (int)tags_l.size(),
                       np, mask, (int)mode, counts.data_ptr<int32_t>());
  }
  auto offs = torch::cumsum(counts, 0, torch::kInt32) - counts;
  int64_t total = np > 0 ? counts.sum().item<int64_t>() : 0;
  auto lidx = torch::empty({total}, opts64);
  bool emit_ridx = mode < 2;
  auto ridx = torch::empty({emit_ridx ? total : 0}, opts64);
  auto matched = torch::zeros({std::max<int64_t>(nb, 1)},
                              torch::dtype(torch::kBool).device(dev));
  if (np > 0 && total > 0) {
    hipLaunchKernelGGL(join_fill_kernel, dim3(grid_1d(np, block)),
                       dim3(block), 0, cur_stream(),
                       table.data_ptr<int64_t>(), next.data_ptr<int64_t>(),
                       (const uint64_t*)probe_hashes.data_ptr<int64_t>(),
                       (const uint64_t*)build_hashes.data_ptr<int64_t>(),
                       (const ColDesc*)pdescs.data_ptr(),
                       (const ColDesc*)bdescs.data_ptr(), (int)tags_l.size(),
                       np, mask, (int)mode, offs.data_ptr<int32_t>(),
                       lidx.data_ptr<int64_t>(),
                       emit_ridx ? ridx.data_ptr<int64_t>() : nullptr,
                       matched.data_ptr<bool>());
  }
  return {lidx, ridx, matched.slice(0, 0, nb)};
}

// ---------------------------------------------------------------------------
// partition helper
// ---------------------------------------------------------------------------

__global__ void u64_mod_kernel(const uint64_t* h, int64_t n, uint64_t d,
                               int64_t* out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (int64_t)(h[i] % d);
}

Tensor u64_mod(Tensor hashes, int64_t n_partitions) {
  int64_t n = hashes.numel();
  auto out = torch::empty({n}, hashes.options());
  if (n > 0) {
    int block = 256;
    hipLaunchKernelGGL(u64_mod_kernel, dim3(grid_1d(n, block)), dim3(block),
                       0, cur_stream(),
                       (const uint64_t*)hashes.data_ptr<int64_t>(), n,
                       (uint64_t)n_partitions, out.data_ptr<int64_t>());
  }
  return out;
}

// ---------------------------------------------------------------------------
// fused grouped multi-aggregate: ONE pass over (gids, value columns)
// computing every sum/min/max/count of an aggregation at once.  The
// per-agg path reads gids once per aggregate (600M x 8B x n_aggs on a
// Q1-shaped plan); here gids load once per row and all accumulators live
// in LDS.  Capability of the reference's per-type agg kernels
// (daft-core/src/array/ops/{sum,mean,count,compare_agg}.rs) fused the
// MI355X way.  ops: 0=sum(f64) 1=min 2=max 3=count-only.
// ---------------------------------------------------------------------------
struct MAggCol {
  const double* data;
  const bool* valid;
  int64_t op;
};

__global__ void grouped_multi_agg_kernel(const int64_t* gids, int64_t n,
                                         int num_groups, int n_aggs,
                                         const MAggCol* cols, double* out,
                                         int64_t* cnt) {
  extern __shared__ char smem[];
  int slots = num_groups * n_aggs;
  double* lacc = (double*)smem;
  uint32_t* lcnt = (uint32_t*)(smem + (size_t)slots * sizeof(double));
  for (int s = threadIdx.x; s < slots; s += blockDim.x) {
    int64_t op = cols[s / num_groups].op;
    if (op == 1)
      ((uint64_t*)lacc)[s] = ~0ull;       // +inf in order-bits
    else if (op == 2)
      ((uint64_t*)lacc)[s] = 0ull;        // -inf in order-bits
    else
      lacc[s] = 0.0;
    lcnt[s] = 0;
  }
  __syncthreads();
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int g = (int)gids[i];
    for (int a = 0; a < n_aggs; ++a) {
      MAggCol c = cols[a];
      if (c.valid && !c.valid[i]) continue;
      int s = a * num_groups + g;
      atomicAdd(&lcnt[s], 1u);
      if (c.op == 0) {
        atomicAdd(&lacc[s], c.data[i]);
      } else if (c.op == 1) {
        atomicMin((unsigned long long*)&lacc[s],
                  (unsigned long long)f64_order_bits(c.data[i]));
      } else if (c.op == 2) {
        atomicMax((unsigned long long*)&lacc[s],
                  (unsigned long long)f64_order_bits(c.data[i]));
      }
    }
  }
  __syncthreads();
  for (int s = threadIdx.x; s < slots; s += blockDim.x) {
    if (lcnt[s] == 0) continue;
    int64_t op = cols[s / num_groups].op;
    atomicAdd((unsigned long long*)&cnt[s], (unsigned long long)lcnt[s]);
    if (op == 0)
      atomicAdd(&out[s], lacc[s]);
    else if (op == 1)
      atomicMin((unsigned long long*)&out[s], ((uint64_t*)lacc)[s]);
    else if (op == 2)
      atomicMax((unsigned long long*)&out[s], ((uint64_t*)lacc)[s]);
  }
}

__global__ void multi_agg_decode_kernel(const MAggCol* cols, int num_groups,
                                        int n_aggs, double* out,
                                        const int64_t* cnt) {
  int s = blockIdx.x * blockDim.x + threadIdx.x;
  if (s >= num_groups * n_aggs) return;
  int64_t op = cols[s / num_groups].op;
  if (op == 1 || op == 2) {
    double v = cnt[s] > 0 ? f64_from_order_bits(((uint64_t*)out)[s]) : 0.0;
    __builtin_memcpy(&out[s], &v, 8);
  }
}

std::vector<Tensor> grouped_multi_agg(Tensor gids, int64_t num_groups,
                                      std::vector<Tensor> datas,
                                      std::vector<OptTensor> valids,
                                      std::vector<int64_t> ops) {
  auto dev = gids.device();
  int n_aggs = (int)ops.size();
  int64_t n = gids.numel();
  int64_t slots = num_groups * n_aggs;
  TORCH_CHECK(slots <= 2048, "grouped_multi_agg: too many slots");
  auto host = torch::empty({n_aggs * 3}, torch::dtype(torch::kInt64));
  int64_t* h = host.data_ptr<int64_t>();
  for (int i = 0; i < n_aggs; ++i) {
    h[i * 3 + 0] = datas[i].defined() && datas[i].numel()
                       ? (int64_t)datas[i].data_ptr<double>() : 0;
    h[i * 3 + 1] = valids[i].has_value()
                       ? (int64_t)valids[i]->data_ptr<bool>() : 0;
    h[i * 3 + 2] = ops[i];
  }
  auto descs = host.to(dev);
  auto out = torch::zeros({slots}, torch::dtype(torch::kFloat64).device(dev));
  // min/max identities as order bits
  {
    auto ob = out.view(torch::kInt64);
    for (int i = 0; i < n_aggs; ++i) {
      if (ops[i] == 1)
        ob.slice(0, i * num_groups, (i + 1) * num_groups).fill_(-1);
      else if (ops[i] == 2)
        ob.slice(0, i * num_groups, (i + 1) * num_groups).fill_(0);
    }
  }
  auto cnt = torch::zeros({slots}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) {
    int block = 256;
    size_t lds = (size_t)slots * (sizeof(double) + sizeof(uint32_t));
    hipLaunchKernelGGL(grouped_multi_agg_kernel,
                       dim3(grid_1d(n, block, 8)), dim3(block), lds,
                       cur_stream(), gids.data_ptr<int64_t>(), n,
                       (int)num_groups, n_aggs,
                       (const MAggCol*)descs.data_ptr(),
                       out.data_ptr<double>(), cnt.data_ptr<int64_t>());
    hipLaunchKernelGGL(multi_agg_decode_kernel,
                       dim3((int)((slots + 255) / 256)), dim3(256), 0,
                       cur_stream(), (const MAggCol*)descs.data_ptr(),
                       (int)num_groups, n_aggs, out.data_ptr<double>(),
                       cnt.data_ptr<int64_t>());
  }
  return {out, cnt};
}

// global-memory variant for large group counts (num_groups beyond LDS):
// still one pass over gids+values for every aggregate — at >>2048 groups
// atomic contention per slot is negligible, the win is reading gids once
// instead of once per aggregate.
__global__ void grouped_multi_agg_global_kernel(const int64_t* gids,
                                                int64_t n,
                                                int64_t num_groups,
                                                int n_aggs,
                                                const MAggCol* cols,
                                                double* out, int64_t* cnt) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t g = gids[i];
    for (int a = 0; a < n_aggs; ++a) {
      MAggCol c = cols[a];
      if (c.valid && !c.valid[i]) continue;
      int64_t s = (int64_t)a * num_groups + g;
      atomicAdd((unsigned long long*)&cnt[s], 1ull);
      if (c.op == 0) {
        atomicAdd(&out[s], c.data[i]);
      } else if (c.op == 1) {
        atomicMin((unsigned long long*)&out[s],
                  (unsigned long long)f64_order_bits(c.data[i]));
      } else if (c.op == 2) {
        atomicMax((unsigned long long*)&out[s],
                  (unsigned long long)f64_order_bits(c.data[i]));
      }
    }
  }
}

__global__ void multi_agg_decode_big_kernel(const MAggCol* cols,
                                            int64_t num_groups, int n_aggs,
                                            double* out,
                                            const int64_t* cnt) {
  int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = num_groups * n_aggs;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; s < total; s += stride) {
    int64_t op = cols[s / num_groups].op;
    if (op == 1 || op == 2) {
      double v = cnt[s] > 0 ? f64_from_order_bits(((uint64_t*)out)[s]) : 0.0;
      __builtin_memcpy(&out[s], &v, 8);
    }
  }
}

std::vector<Tensor> grouped_multi_agg_big(Tensor gids, int64_t num_groups,
                                          std::vector<Tensor> datas,
                                          std::vector<OptTensor> valids,
                                          std::vector<int64_t> ops) {
  auto dev = gids.device();
  int n_aggs = (int)ops.size();
  int64_t n = gids.numel();
  int64_t slots = num_groups * n_aggs;
  auto host = torch::empty({n_aggs * 3}, torch::dtype(torch::kInt64));
  int64_t* h = host.data_ptr<int64_t>();
  for (int i = 0; i < n_aggs; ++i) {
    h[i * 3 + 0] = datas[i].defined() && datas[i].numel()
                       ? (int64_t)datas[i].data_ptr<double>() : 0;
    h[i * 3 + 1] = valids[i].has_value()
                       ? (int64_t)valids[i]->data_ptr<bool>() : 0;
    h[i * 3 + 2] = ops[i];
  }
  auto descs = host.to(dev);
  auto out = torch::zeros({slots}, torch::dtype(torch::kFloat64).device(dev));
  {
    auto ob = out.view(torch::kInt64);
    for (int i = 0; i < n_aggs; ++i) {
      if (ops[i] == 1)
        ob.slice(0, i * num_groups, (i + 1) * num_groups).fill_(-1);
      else if (ops[i] == 2)
        ob.slice(0, i * num_groups, (i + 1) * num_groups).fill_(0);
    }
  }
  auto cnt = torch::zeros({slots}, torch::dtype(torch::kInt64).device(dev));
  if (n > 0) {
    int block = 256;
    hipLaunchKernelGGL(grouped_multi_agg_global_kernel,
                       dim3(grid_1d(n, block, 4)), dim3(block), 0,
                       cur_stream(), gids.data_ptr<int64_t>(), n,
                       num_groups, n_aggs,
                       (const MAggCol*)descs.data_ptr(),
                       out.data_ptr<double>(), cnt.data_ptr<int64_t>());
    hipLaunchKernelGGL(multi_agg_decode_big_kernel,
                       dim3(grid_1d(slots, 256)), dim3(256), 0,
                       cur_stream(), (const MAggCol*)descs.data_ptr(),
                       num_groups, n_aggs, out.data_ptr<double>(),
                       cnt.data_ptr<int64_t>());
  }
  return {out, cnt};
}
