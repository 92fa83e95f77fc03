// HIP runtime presence probe. Compiled with hipcc; links libamdhip64.
#include <hip/hip_runtime.h>

namespace hipflux {

int hip_device_count() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

}  // namespace hipflux
