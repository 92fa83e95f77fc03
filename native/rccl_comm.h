// RCCL-over-xGMI collective layer for tile-parallel encode
// (BASELINE config 5's communication pattern).
//
// xGMI is point-to-point (7 links/GPU): the latency-critical per-frame
// tile-boundary exchange uses a ONE-HOP schedule (grouped ncclSend/
// ncclRecv to every peer — each payload crosses exactly one link)
// rather than a ring all-gather, which serializes small payloads over
// per-link hops. Ring (ncclAllGather) remains for bulk reference-frame
// sharing where bandwidth, not latency, dominates.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace hipflux {

class TileComm {
 public:
  // uid: the ncclUniqueId bytes from make_uid() on rank 0, distributed
  // out-of-band (the bench uses the torch.distributed store).
  TileComm(int rank, int world, const std::string& uid, int device);
  ~TileComm();

  static std::string make_uid();

  int rank() const { return rank_; }
  int world() const { return world_; }

  // All-gather `bytes` from src_dev (device pointer; this rank's slot)
  // so every rank holds every rank's payload. schedule: 0 = ring
  // (ncclAllGather), 1 = one-hop p2p. Returns elapsed ms (stream-synced).
  double exchange(uintptr_t src_dev, size_t bytes, int schedule);

  // Broadcast a reference-frame-sized buffer from root (ring/tree inside
  // RCCL; bulk path). Returns elapsed ms.
  double broadcast(uintptr_t buf_dev, size_t bytes, int root);

  // D2H of one rank's gathered slot (tests / halo consumers).
  std::vector<uint8_t> gathered(int rank, size_t bytes);

 private:
  void ensure(size_t bytes);

  void* comm_ = nullptr;          // ncclComm_t
  void* stream_ = nullptr;        // hipStream_t
  void* ev0_ = nullptr;           // hipEvent_t
  void* ev1_ = nullptr;
  void* d_all_ = nullptr;         // world * slot bytes
  size_t slot_cap_ = 0;
  size_t last_bytes_ = 0;
  int rank_ = 0, world_ = 1, device_ = 0;
};

}  // namespace hipflux
