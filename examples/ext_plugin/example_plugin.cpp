// Example daft_amd extension plugin (see daft_amd/ext/daft_ext.h).
// Build:
//   g++ -O2 -shared -fPIC -I<repo>/daft_amd/ext example_plugin.cpp \
//       -o example_plugin.so
// (hipcc works the same way for plugins that launch gfx950 kernels on
// device-resident columns: check col.device == 1 and use the pointers
// as HBM addresses.)
//
// Use:
//   import daft_amd as daft
//   daft.load_extension("example_plugin.so")
//   df.select(daft.ext_function("ext_add1", col("a")))
#include <cstdio>
#include <cstring>
#include <cmath>
#include "daft_ext.h"

static int add1(const DaftExtColumn* args, int32_t n_args,
                DaftExtColumn* out, char* err, int32_t err_len) {
  if (n_args != 1 || args[0].dtype != DAFT_EXT_INT64) {
    snprintf(err, err_len, "ext_add1 expects one int64 column");
    return 1;
  }
  if (args[0].device != 0) {
    snprintf(err, err_len, "ext_add1 is host-only in this example");
    return 1;
  }
  const int64_t* in = (const int64_t*)args[0].data;
  int64_t* o = (int64_t*)out->data;
  for (int64_t i = 0; i < args[0].length; ++i) o[i] = in[i] + 1;
  if (args[0].validity && out->validity)
    memcpy(out->validity, args[0].validity, (size_t)args[0].length);
  return 0;
}

static int hypot_fn(const DaftExtColumn* args, int32_t n_args,
                    DaftExtColumn* out, char* err, int32_t err_len) {
  if (n_args != 2 || args[0].dtype != DAFT_EXT_FLOAT64 ||
      args[1].dtype != DAFT_EXT_FLOAT64) {
    snprintf(err, err_len, "ext_hypot expects two float64 columns");
    return 1;
  }
  const double* a = (const double*)args[0].data;
  const double* b = (const double*)args[1].data;
  double* o = (double*)out->data;
  for (int64_t i = 0; i < args[0].length; ++i) o[i] = std::hypot(a[i], b[i]);
  return 0;
}

extern "C" int daft_ext_abi_version(void) { return DAFT_EXT_ABI_V1; }

extern "C" int daft_ext_register(DaftExtApi* api) {
  if (api->abi_version != DAFT_EXT_ABI_V1) return 1;
  api->register_scalar(api->host, "ext_add1", DAFT_EXT_INT64, add1);
  api->register_scalar(api->host, "ext_hypot", DAFT_EXT_FLOAT64, hypot_fn);
  return 0;
}
