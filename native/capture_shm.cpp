// Shared-memory frame source: the bridge from the in-tree Wayland
// compositor (selkies_amd/wayland/compositor.py) — or any producer —
// into the native engine's capture loop, without a Python round trip
// per frame on the native side.
//
// File layout at `path` (little endian, producer-created):
//   u32 magic   'HFSH' (0x48534648)
//   u32 seq     even = stable; producer writes seq+1 (odd), pixels,
//               then seq+2 — the classic seqlock, so acquire() never
//               returns a torn frame
//   u32 width
//   u32 height
//   u8  pixels[width*height*4]   BGRX
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <thread>
#include <vector>

#include "capture.h"

namespace hipflux {
namespace {

constexpr uint32_t kShmMagic = 0x48534648;  // 'HFSH'

struct ShmHeader {
  uint32_t magic;
  uint32_t seq;
  uint32_t width;
  uint32_t height;
};

class ShmSource : public FrameSource {
 public:
  ShmSource(void* map, size_t len, int w, int h)
      : map_(map), len_(len), w_(w), h_(h) {
    buf_.resize(static_cast<size_t>(w_) * h_ * 4);
  }

  ~ShmSource() override {
    if (map_) ::munmap(map_, len_);
  }

  bool acquire(RawFrame& out) override {
    const auto* hdr = static_cast<const ShmHeader*>(map_);
    const uint8_t* px = static_cast<const uint8_t*>(map_) +
                        sizeof(ShmHeader);
    // seqlock read: retry while the producer is mid-write or raced us
    for (int attempt = 0; attempt < 64; ++attempt) {
      uint32_t s0 = __atomic_load_n(&hdr->seq, __ATOMIC_ACQUIRE);
      if (s0 & 1) {
        std::this_thread::sleep_for(std::chrono::microseconds(200));
        continue;
      }
      std::memcpy(buf_.data(), px, buf_.size());
      uint32_t s1 = __atomic_load_n(&hdr->seq, __ATOMIC_ACQUIRE);
      if (s0 == s1) break;
    }
    out.data = buf_.data();
    out.width = w_;
    out.height = h_;
    out.stride = w_ * 4;
    out.ts_ms = now_ms();
    return true;
  }

  int width() const override { return w_; }
  int height() const override { return h_; }

 private:
  void* map_;
  size_t len_;
  int w_, h_;
  std::vector<uint8_t> buf_;
};

}  // namespace

std::unique_ptr<FrameSource> make_shm_source(const std::string& path) {
  int fd = ::open(path.c_str(), O_RDONLY);
  if (fd < 0) {
    std::fprintf(stderr, "hipflux: shm source: cannot open %s\n",
                 path.c_str());
    return nullptr;
  }
  struct stat st{};
  if (::fstat(fd, &st) != 0 ||
      st.st_size < static_cast<off_t>(sizeof(ShmHeader))) {
    ::close(fd);
    return nullptr;
  }
  void* map = ::mmap(nullptr, st.st_size, PROT_READ, MAP_SHARED, fd, 0);
  ::close(fd);
  if (map == MAP_FAILED) return nullptr;
  const auto* hdr = static_cast<const ShmHeader*>(map);
  if (hdr->magic != kShmMagic || hdr->width == 0 || hdr->height == 0 ||
      static_cast<size_t>(st.st_size) <
          sizeof(ShmHeader) +
              static_cast<size_t>(hdr->width) * hdr->height * 4) {
    std::fprintf(stderr, "hipflux: shm source: bad header in %s\n",
                 path.c_str());
    ::munmap(map, st.st_size);
    return nullptr;
  }
  return std::make_unique<ShmSource>(map, st.st_size, hdr->width,
                                     hdr->height);
}

}  // namespace hipflux
