// Temporary stubs for pipelines not yet implemented. Each returns nullptr so
// the engine can fall back; replaced as the real implementations land.
#include <cstdio>

#include "engine.h"

namespace hipflux {

#ifndef HIPFLUX_HAVE_H264
std::unique_ptr<EncodePipeline> make_cpu_h264_pipeline(
    const CaptureSettings&) {
  std::fprintf(stderr, "hipflux: CPU H.264 pipeline not built yet\n");
  return nullptr;
}
#endif

#ifndef HIPFLUX_HAVE_HIP
std::unique_ptr<EncodePipeline> make_hip_pipeline(const CaptureSettings&) {
  return nullptr;
}
#endif

}  // namespace hipflux
