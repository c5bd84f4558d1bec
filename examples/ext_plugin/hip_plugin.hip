// GPU extension plugin: launches its own gfx950 kernel directly on the
// engine's HBM-resident column buffers (no copies at the ABI boundary).
// Build on an MI355X box:
//   hipcc --offload-arch=gfx950 -O2 -shared -fPIC \
//       -I<repo>/daft_amd/ext hip_plugin.hip -o hip_plugin.so
#include <cstdio>
#include <hip/hip_runtime.h>
#include "daft_ext.h"

__global__ void saxpy_kernel(const double* x, double* out, int64_t n,
                             double a, double b) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = a * x[i] + b;
}

static int ext_saxpy(const DaftExtColumn* args, int32_t n_args,
                     DaftExtColumn* out, char* err, int32_t err_len) {
  if (n_args != 1 || args[0].dtype != DAFT_EXT_FLOAT64) {
    snprintf(err, err_len, "ext_saxpy expects one float64 column");
    return 1;
  }
  int64_t n = args[0].length;
  if (args[0].device == 1) {
    int block = 256;
    int grid = (int)((n + block * 4 - 1) / ((int64_t)block * 4));
    if (grid < 1) grid = 1;
    if (grid > 65535) grid = 65535;
    hipLaunchKernelGGL(saxpy_kernel, dim3(grid), dim3(block), 0, 0,
                       (const double*)args[0].data, (double*)out->data, n,
                       2.0, 1.0);
    hipError_t e = hipGetLastError();
    if (e != hipSuccess) {
      snprintf(err, err_len, "hip: %s", hipGetErrorString(e));
      return 1;
    }
    return 0;
  }
  const double* x = (const double*)args[0].data;
  double* o = (double*)out->data;
  for (int64_t i = 0; i < n; ++i) o[i] = 2.0 * x[i] + 1.0;
  return 0;
}

extern "C" int daft_ext_abi_version(void) { return DAFT_EXT_ABI_V1; }

extern "C" int daft_ext_register(DaftExtApi* api) {
  if (api->abi_version != DAFT_EXT_ABI_V1) return 1;
  api->register_scalar(api->host, "ext_saxpy", DAFT_EXT_FLOAT64,
                       ext_saxpy);
  return 0;
}
