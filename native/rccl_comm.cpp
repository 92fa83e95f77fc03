// RCCL collective layer implementation (see rccl_comm.h).
#include "rccl_comm.h"

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <cstring>
#include <stdexcept>

namespace hipflux {

#define HIP_CK(expr)                                                       \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("HIP: ") +                      \
                               hipGetErrorString(_e) + " at " #expr);      \
  } while (0)

#define NCCL_CK(expr)                                                      \
  do {                                                                     \
    ncclResult_t _r = (expr);                                              \
    if (_r != ncclSuccess)                                                 \
      throw std::runtime_error(std::string("RCCL: ") +                     \
                               ncclGetErrorString(_r) + " at " #expr);     \
  } while (0)

std::string TileComm::make_uid() {
  ncclUniqueId id;
  NCCL_CK(ncclGetUniqueId(&id));
  return std::string(id.internal, id.internal + NCCL_UNIQUE_ID_BYTES);
}

TileComm::TileComm(int rank, int world, const std::string& uid, int device)
    : rank_(rank), world_(world), device_(device) {
  if (uid.size() != NCCL_UNIQUE_ID_BYTES)
    throw std::runtime_error("bad RCCL unique id size");
  HIP_CK(hipSetDevice(device));
  hipStream_t s;
  HIP_CK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  stream_ = s;
  hipEvent_t e0, e1;
  HIP_CK(hipEventCreate(&e0));
  HIP_CK(hipEventCreate(&e1));
  ev0_ = e0;
  ev1_ = e1;
  ncclUniqueId id;
  std::memcpy(id.internal, uid.data(), NCCL_UNIQUE_ID_BYTES);
  ncclComm_t c;
  NCCL_CK(ncclCommInitRank(&c, world, id, rank));
  comm_ = c;
}

TileComm::~TileComm() {
  if (comm_) (void)ncclCommDestroy(static_cast<ncclComm_t>(comm_));
  if (d_all_) (void)hipFree(d_all_);
  if (ev0_) (void)hipEventDestroy(static_cast<hipEvent_t>(ev0_));
  if (ev1_) (void)hipEventDestroy(static_cast<hipEvent_t>(ev1_));
  if (stream_) (void)hipStreamDestroy(static_cast<hipStream_t>(stream_));
}

void TileComm::ensure(size_t bytes) {
  if (bytes <= slot_cap_) return;
  if (d_all_) (void)hipFree(d_all_);
  HIP_CK(hipMalloc(&d_all_, bytes * world_));
  slot_cap_ = bytes;
}

double TileComm::exchange(uintptr_t src_dev, size_t bytes, int schedule) {
  ensure(bytes);
  last_bytes_ = bytes;
  auto s = static_cast<hipStream_t>(stream_);
  auto c = static_cast<ncclComm_t>(comm_);
  auto* src = reinterpret_cast<void*>(src_dev);
  auto* all = static_cast<uint8_t*>(d_all_);
  HIP_CK(hipEventRecord(static_cast<hipEvent_t>(ev0_), s));
  if (schedule == 0) {
    NCCL_CK(ncclAllGather(src, all, bytes, ncclUint8, c, s));
  } else {
    // one-hop everywhere: each peer pair exchanges directly over its
    // xGMI link; no multi-hop ring latency for small boundary payloads
    NCCL_CK(ncclGroupStart());
    for (int p = 0; p < world_; ++p) {
      if (p == rank_) continue;
      NCCL_CK(ncclSend(src, bytes, ncclUint8, p, c, s));
      NCCL_CK(ncclRecv(all + static_cast<size_t>(p) * bytes, bytes,
                       ncclUint8, p, c, s));
    }
    NCCL_CK(ncclGroupEnd());
    HIP_CK(hipMemcpyAsync(all + static_cast<size_t>(rank_) * bytes, src,
                          bytes, hipMemcpyDeviceToDevice, s));
  }
  HIP_CK(hipEventRecord(static_cast<hipEvent_t>(ev1_), s));
  HIP_CK(hipEventSynchronize(static_cast<hipEvent_t>(ev1_)));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, static_cast<hipEvent_t>(ev0_),
                            static_cast<hipEvent_t>(ev1_));
  return ms;
}

double TileComm::broadcast(uintptr_t buf_dev, size_t bytes, int root) {
  auto s = static_cast<hipStream_t>(stream_);
  HIP_CK(hipEventRecord(static_cast<hipEvent_t>(ev0_), s));
  NCCL_CK(ncclBroadcast(reinterpret_cast<void*>(buf_dev),
                        reinterpret_cast<void*>(buf_dev), bytes, ncclUint8,
                        root, static_cast<ncclComm_t>(comm_), s));
  HIP_CK(hipEventRecord(static_cast<hipEvent_t>(ev1_), s));
  HIP_CK(hipEventSynchronize(static_cast<hipEvent_t>(ev1_)));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, static_cast<hipEvent_t>(ev0_),
                            static_cast<hipEvent_t>(ev1_));
  return ms;
}

std::vector<uint8_t> TileComm::gathered(int rank, size_t bytes) {
  std::vector<uint8_t> out(bytes);
  if (!d_all_ || bytes > last_bytes_)
    throw std::runtime_error("no gathered payload of that size");
  HIP_CK(hipMemcpy(out.data(),
                   static_cast<uint8_t*>(d_all_) +
                       static_cast<size_t>(rank) * last_bytes_,
                   bytes, hipMemcpyDeviceToHost));
  return out;
}

}  // namespace hipflux
