// Generic multi-column row descriptors for hashing / equality on device.
// The GPU analog of the reference's typed row comparators
// (daft-recordbatch/src/probeable/probes.rs): a small descriptor array in
// device memory describes each key column; row_hash/row_eq loop over it.
#pragma once

#include "common.h"

// tags shared with daft_amd/kernels/__init__.py
enum ColTag : int32_t {
  TAG_W1 = 0,
  TAG_W2 = 1,
  TAG_W4 = 2,
  TAG_W8 = 3,
  TAG_STR = 4,
  TAG_F32 = 5,
  TAG_F64 = 6,
};

struct ColDesc {
  const void* data;       // fixed-width values or utf8 bytes
  const int64_t* offsets; // strings only
  const bool* validity;   // may be null
  int64_t tag;
};

constexpr uint64_t kNullHash = 0x9E3779B97F4A7C15ull;

DEV_INLINE uint64_t canon_f64_bits(double v) {
  if (v == 0.0) v = 0.0;            // -0.0 -> +0.0
  if (v != v) return 0x7FF8000000000000ull;  // canonical NaN
  uint64_t b;
  __builtin_memcpy(&b, &v, 8);
  return b;
}

DEV_INLINE uint64_t col_value_hash(const ColDesc& c, int64_t i) {
  if (c.validity && !c.validity[i]) return kNullHash;
  switch (c.tag) {
    case TAG_W1:
      return splitmix64((uint64_t)((const uint8_t*)c.data)[i]);
    case TAG_W2:
      return splitmix64((uint64_t)((const uint16_t*)c.data)[i]);
    case TAG_W4:
      return splitmix64((uint64_t)((const uint32_t*)c.data)[i]);
    case TAG_W8:
      return splitmix64(((const uint64_t*)c.data)[i]);
    case TAG_F32:
      return splitmix64(canon_f64_bits((double)((const float*)c.data)[i]));
    case TAG_F64:
      return splitmix64(canon_f64_bits(((const double*)c.data)[i]));
    case TAG_STR: {
      int64_t a = c.offsets[i], b = c.offsets[i + 1];
      return hash_bytes_dev((const uint8_t*)c.data + a, b - a);
    }
  }
  return 0;
}

DEV_INLINE uint64_t row_hash(const ColDesc* cols, int ncols, int64_t i,
                             uint64_t seed) {
  uint64_t acc = seed + 0x8445D61A4E774912ull;
  for (int c = 0; c < ncols; ++c) {
    acc = (acc * 0x9E3779B97F4A7C15ull) ^ splitmix64(col_value_hash(cols[c], i));
  }
  return acc;
}

// SQL-style equality for grouping: null == null is TRUE inside a groupby
// (rows with null keys form one group) but FALSE for joins.  `null_eq`
// selects the semantic.
DEV_INLINE bool col_value_eq(const ColDesc& a, int64_t i, const ColDesc& b,
                             int64_t j, bool null_eq) {
  bool va = !a.validity || a.validity[i];
  bool vb = !b.validity || b.validity[j];
  if (!va || !vb) return null_eq && !va && !vb;
  switch (a.tag) {
    case TAG_W1:
      return ((const uint8_t*)a.data)[i] == ((const uint8_t*)b.data)[j];
    case TAG_W2:
      return ((const uint16_t*)a.data)[i] == ((const uint16_t*)b.data)[j];
    case TAG_W4:
      return ((const uint32_t*)a.data)[i] == ((const uint32_t*)b.data)[j];
    case TAG_W8:
      return ((const uint64_t*)a.data)[i] == ((const uint64_t*)b.data)[j];
    case TAG_F32:
      return canon_f64_bits((double)((const float*)a.data)[i]) ==
             canon_f64_bits((double)((const float*)b.data)[j]);
    case TAG_F64:
      return canon_f64_bits(((const double*)a.data)[i]) ==
             canon_f64_bits(((const double*)b.data)[j]);
    case TAG_STR: {
      int64_t sa = a.offsets[i], ea = a.offsets[i + 1];
      int64_t sb = b.offsets[j], eb = b.offsets[j + 1];
      int64_t la = ea - sa, lb = eb - sb;
      if (la != lb) return false;
      const uint8_t* pa = (const uint8_t*)a.data + sa;
      const uint8_t* pb = (const uint8_t*)b.data + sb;
      int64_t k = 0;
      for (; k + 8 <= la; k += 8) {
        uint64_t wa, wb;
        __builtin_memcpy(&wa, pa + k, 8);
        __builtin_memcpy(&wb, pb + k, 8);
        if (wa != wb) return false;
      }
      for (; k < la; ++k)
        if (pa[k] != pb[k]) return false;
      return true;
    }
  }
  return false;
}

DEV_INLINE bool row_eq(const ColDesc* a, const ColDesc* b, int ncols,
                       int64_t i, int64_t j, bool null_eq) {
  for (int c = 0; c < ncols; ++c)
    if (!col_value_eq(a[c], i, b[c], j, null_eq)) return false;
  return true;
}

DEV_INLINE bool row_has_null(const ColDesc* cols, int ncols, int64_t i) {
  for (int c = 0; c < ncols; ++c)
    if (cols[c].validity && !cols[c].validity[i]) return true;
  return false;
}
