// XgmiMesh: the intra-node device-side transport for the hip_* algorithms.
//
// This is the MI355X-native replacement for the reference's CUDA-side
// transport plumbing (gloo/cuda_workspace.h workspaces + bound-buffer
// rings over TCP): instead of staging GPU chunks through host memory and
// sockets, every rank exports one coarse-grained HBM scratch region
// (work buffer + double-buffered inboxes) and one fine-grained flag page
// via hipIpcMemHandle; peers map them and move chunks directly
// GPU-to-GPU with hipMemcpyAsync over xGMI, publishing progress through
// stream-ordered system-scope doorbell kernels (csrc/hip/kernels.hip).
// The host only enqueues: the entire ring/halving-doubling schedule runs
// on two HIP streams with GPU-side flag waits, so per-step host latency
// (the reference's waitRecv/copyAsync round trips, cuda_allreduce_ring_
// chunked.cc:130-273) disappears from the critical path.
//
// Construction is collective over a connected glooamd::Context: IPC
// handles travel over the context's TCP pairs (v2 allgather).
#pragma once

#include <hip/hip_runtime.h>

#include <memory>
#include <vector>

#include "context.h"
#include "hip/core.h"

namespace glooamd {
namespace hip {

class XgmiMesh {
 public:
  // workCap: bytes for the working copy of user data.
  // inboxCap: bytes per inbox (x2 inboxes, double buffered).
  XgmiMesh(
      std::shared_ptr<Context> ctx,
      int device,
      size_t workCap,
      size_t inboxCap);
  ~XgmiMesh();
  XgmiMesh(const XgmiMesh&) = delete;

  int device() const {
    return device_;
  }
  size_t workCap() const {
    return workCap_;
  }
  size_t inboxCap() const {
    return inboxCap_;
  }
  // Grow the scratch region (collective; invalidates peer pointers and
  // re-exchanges handles).
  void ensureCapacity(size_t workCap, size_t inboxCap);

  char* work() {
    return data_;
  }
  char* inbox(int idx) {
    return data_ + workCap_ + static_cast<size_t>(idx) * inboxCap_;
  }
  char* peerWork(int rank) {
    return peerData_[rank];
  }
  char* peerInbox(int rank, int idx) {
    return peerData_[rank] + workCap_ + static_cast<size_t>(idx) * inboxCap_;
  }

  static constexpr int kNumFlags = 512;
  uint64_t* flag(int idx) {
    return flags_ + idx;
  }
  uint64_t* peerFlag(int rank, int idx) {
    return peerFlags_[rank] + idx;
  }
  // Bump-allocate flag indices; collective construction order keeps the
  // allocation identical on every rank.
  int allocFlags(int count);

  // Host-side poison: release every local GPU-side flag wait (used by the
  // run() watchdog on timeout so streams can drain before throwing).
  void poisonFlags();

  // Debug: host copy of the first n local flag values.
  std::vector<uint64_t> readFlags(int n = 16);

 private:
  void exchange(); // allgather + open peer handles
  void releasePeers();

  std::shared_ptr<Context> ctx_;
  int device_;
  size_t workCap_;
  size_t inboxCap_;

  char* data_{nullptr}; // coarse-grained: work + 2 inboxes
  std::vector<char*> retired_; // outgrown buffers (freed at destruction)
  uint64_t* flags_{nullptr}; // fine-grained page, peer-writable
  std::vector<char*> peerData_;
  std::vector<uint64_t*> peerFlags_;
  std::vector<bool> peerSameProcess_;
  int nextFlag_{0};
};

} // namespace hip
} // namespace glooamd
