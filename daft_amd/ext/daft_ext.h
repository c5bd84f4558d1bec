/* daft_amd extension ABI (capability of the reference's stable C-ABI
 * plugin system, /root/reference/src/daft-ext/src/abi/ +
 * ffi/trampoline.rs — third-party .so extensions loaded at runtime).
 *
 * MI355X-native design: columns are exposed as raw device/host buffers
 * (the engine's Series are torch tensors in Arrow layouts), so a plugin
 * built with hipcc can launch its own gfx950 kernels directly on column
 * memory — no serialization at the boundary.
 *
 * A plugin is a shared library exporting:
 *     int daft_ext_abi_version(void);       // must return DAFT_EXT_ABI_V1
 *     int daft_ext_register(DaftExtApi*);   // 0 on success
 */
#ifndef DAFT_EXT_H
#define DAFT_EXT_H
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define DAFT_EXT_ABI_V1 1

/* dtype codes for DaftExtColumn.dtype */
enum {
  DAFT_EXT_INT64 = 0,
  DAFT_EXT_FLOAT64 = 1,
  DAFT_EXT_FLOAT32 = 2,
  DAFT_EXT_INT32 = 3,
  DAFT_EXT_UINT8 = 4,
  DAFT_EXT_BOOL = 5,
};

typedef struct {
  void* data;       /* fixed-width buffer, length*sizeof(dtype) bytes   */
  int64_t length;   /* number of rows                                   */
  int32_t dtype;    /* DAFT_EXT_* code                                  */
  int32_t device;   /* 0 = host, 1 = GPU (HBM pointer, default stream)  */
  uint8_t* validity;/* 1 byte per row (1 = valid) or NULL = all valid   */
} DaftExtColumn;

/* Scalar function: read `args[0..n_args)`, fill `out` (pre-allocated by
 * the host with out->length == args[0].length and the registered output
 * dtype; out->validity is writable, pre-filled all-valid).  Return 0 on
 * success; on failure write a message into err (err_len bytes) and
 * return nonzero. */
typedef int (*DaftExtScalarFn)(const DaftExtColumn* args, int32_t n_args,
                               DaftExtColumn* out, char* err,
                               int32_t err_len);

typedef struct {
  int32_t abi_version;   /* DAFT_EXT_ABI_V1 */
  void* host;            /* opaque host handle: pass to callbacks */
  /* Register a scalar function under `name` producing `out_dtype`. */
  int (*register_scalar)(void* host, const char* name, int32_t out_dtype,
                         DaftExtScalarFn fn);
} DaftExtApi;

#ifdef __cplusplus
}
#endif
#endif /* DAFT_EXT_H */
