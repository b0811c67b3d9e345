#include "hip/mesh.h"

#include <unistd.h>

#include <cstring>

#include "collectives/collectives.h"
#include "common/logging.h"
#include "common/utils.h"

namespace glooamd {
namespace hip {

namespace {
// hipExtMallocWithFlags with fine-grained coherence for the flag page:
// cross-device system-scope atomics over xGMI require fine-grained
// allocations on CDNA.
void* allocFineGrained(size_t bytes) {
  void* p = nullptr;
  hipError_t err =
      hipExtMallocWithFlags(&p, bytes, hipDeviceMallocFinegrained);
  if (err != hipSuccess || p == nullptr) {
    // Fallback: pinned host memory is always fine-grained & peer-visible
    // (slower polls, still correct).
    GA_WARN << "fine-grained device alloc failed ("
            << hipGetErrorString(err) << "); flags fall back to hipMalloc";
    GA_HIP_CHECK(hipMalloc(&p, bytes));
  }
  return p;
}

// Flag pages are IMMORTAL AND NEVER REUSED: a stale doorbell packet
// (CP wait-value, spin kernel, or an unexecuted flag write left on a
// quarantined stream of a failed, poisoned run — possibly in a PEER
// process via the IPC mapping) may still reference the page long after
// its mesh died. Freeing would make that packet touch freed memory;
// recycling would let a late stale WRITE corrupt an unrelated new
// mesh's flags. Leaking is strictly safe: a page is 4 KiB per mesh
// lifetime (a full test-suite run leaks a couple of MB).
std::mutex& flagPoolMutex() {
  static std::mutex mu;
  return mu;
}
uint64_t* acquireFlagPage(size_t count) {
  return static_cast<uint64_t*>(allocFineGrained(count * 8));
}
void releaseFlagPage(uint64_t* p) {
  static auto* graveyard = new std::vector<uint64_t*>();
  std::lock_guard<std::mutex> lock(flagPoolMutex());
  graveyard->push_back(p);
}

struct HandleBlob {
  int32_t pid;
  int32_t device;
  uint64_t dataPtr;
  uint64_t flagsPtr;
  hipIpcMemHandle_t dataHandle;
  hipIpcMemHandle_t flagsHandle;
};
} // namespace

XgmiMesh::XgmiMesh(
    std::shared_ptr<Context> ctx,
    int device,
    size_t workCap,
    size_t inboxCap)
    : ctx_(std::move(ctx)),
      device_(device),
      workCap_(workCap),
      inboxCap_(inboxCap) {
  GA_HIP_CHECK(hipSetDevice(device_));
  const size_t total = workCap_ + 2 * inboxCap_;
  {
    std::lock_guard<std::mutex> lock(allocMutex());
    GA_HIP_CHECK(hipMalloc(reinterpret_cast<void**>(&data_), total));
  }
  flags_ = acquireFlagPage(kNumFlags);
  // hipMemset (sync form) completes before returning; no device-wide
  // sync here (it could block on other ranks' in-flight kernels when
  // several ranks share one GPU).
  GA_HIP_CHECK(hipMemset(flags_, 0, kNumFlags * 8));
  exchange();
}

void XgmiMesh::exchange() {
  const int P = ctx_->size;
  peerData_.assign(P, nullptr);
  peerFlags_.assign(P, nullptr);
  peerSameProcess_.assign(P, false);

  HandleBlob mine;
  std::memset(&mine, 0, sizeof(mine));
  mine.pid = getpid();
  mine.device = device_;
  mine.dataPtr = reinterpret_cast<uint64_t>(data_);
  mine.flagsPtr = reinterpret_cast<uint64_t>(flags_);
  if (P > 1) {
    // Exporting handles only matters with real peers; it also fails for
    // some allocation types, so only do it when needed.
    GA_HIP_CHECK(hipIpcGetMemHandle(&mine.dataHandle, data_));
    GA_HIP_CHECK(hipIpcGetMemHandle(&mine.flagsHandle, flags_));
  }

  std::vector<HandleBlob> all(P);
  AllgatherOptions opts(ctx_);
  opts.input = &mine;
  opts.output = all.data();
  opts.inElements = sizeof(HandleBlob);
  opts.elementSize = 1;
  opts.tag = ctx_->nextSlot();
  allgather(opts);

  for (int r = 0; r < P; r++) {
    if (r == ctx_->rank) {
      peerData_[r] = data_;
      peerFlags_[r] = flags_;
      peerSameProcess_[r] = true;
      continue;
    }
    if (all[r].pid == mine.pid) {
      // Thread-spawned rank in this process: IPC open of a same-process
      // handle is invalid; the raw pointers are directly usable.
      peerData_[r] = reinterpret_cast<char*>(all[r].dataPtr);
      peerFlags_[r] = reinterpret_cast<uint64_t*>(all[r].flagsPtr);
      peerSameProcess_[r] = true;
    } else {
      void* p = nullptr;
      GA_HIP_CHECK(hipIpcOpenMemHandle(
          &p, all[r].dataHandle, hipIpcMemLazyEnablePeerAccess));
      peerData_[r] = static_cast<char*>(p);
      void* f = nullptr;
      GA_HIP_CHECK(hipIpcOpenMemHandle(
          &f, all[r].flagsHandle, hipIpcMemLazyEnablePeerAccess));
      peerFlags_[r] = static_cast<uint64_t*>(f);
    }
  }

  if (getEnvFlag("GLOO_AMD_FLAG_DEBUG")) {
    std::string msg = "mesh exchange rank " + std::to_string(ctx_->rank) +
        " data=" + std::to_string(reinterpret_cast<uintptr_t>(data_)) +
        " flags=" + std::to_string(reinterpret_cast<uintptr_t>(flags_));
    for (int r = 0; r < P; r++) {
      msg += " peer" + std::to_string(r) + "=(" +
          std::to_string(reinterpret_cast<uintptr_t>(peerData_[r])) + "," +
          std::to_string(reinterpret_cast<uintptr_t>(peerFlags_[r])) + ")";
    }
    GA_ERROR << msg;
  }
  // Everyone has mapped everyone before first use.
  BarrierOptions bar(ctx_);
  bar.tag = ctx_->nextSlot();
  barrier(bar);
}

void XgmiMesh::releasePeers() {
  for (int r = 0; r < static_cast<int>(peerData_.size()); r++) {
    if (r == ctx_->rank || peerSameProcess_[r]) {
      continue;
    }
    if (peerData_[r] != nullptr) {
      (void)hipIpcCloseMemHandle(peerData_[r]);
    }
    if (peerFlags_[r] != nullptr) {
      (void)hipIpcCloseMemHandle(peerFlags_[r]);
    }
  }
  peerData_.clear();
  peerFlags_.clear();
}

void XgmiMesh::ensureCapacity(size_t workCap, size_t inboxCap) {
  const bool trace = getEnvFlag("GLOO_AMD_PHASE_TRACE");
  if (workCap <= workCap_ && inboxCap <= inboxCap_) {
    // Still a collective decision: all ranks compute the same sizes from
    // the same collective arguments, so either all grow or none do.
    return;
  }
  // Over-allocate on growth so regrowth is rare.
  workCap_ = std::max(std::max(workCap_ * 2, workCap), size_t(8) << 20);
  inboxCap_ = std::max(inboxCap_, inboxCap);
  // Close every mapping of the old buffers collectively, but do NOT free
  // the old allocation: hipIpcGetMemHandle on an allocation recycled
  // from freed-while-peer-mapped pages fails with invalid argument, so
  // retired buffers live until the mesh is destroyed.
  if (trace) {
    GA_ERROR << "[phase r" << ctx_->rank << "] mesh:grow-barrier1";
  }
  {
    BarrierOptions bar(ctx_);
    bar.tag = ctx_->nextSlot();
    barrier(bar);
  }
  releasePeers();
  if (trace) {
    GA_ERROR << "[phase r" << ctx_->rank << "] mesh:grow-barrier2";
  }
  {
    BarrierOptions bar(ctx_);
    bar.tag = ctx_->nextSlot();
    barrier(bar);
  }
  retired_.push_back(data_);
  data_ = nullptr;
  if (trace) {
    GA_ERROR << "[phase r" << ctx_->rank << "] mesh:grow-malloc";
  }
  {
    std::lock_guard<std::mutex> lock(allocMutex());
    GA_HIP_CHECK(hipMalloc(
        reinterpret_cast<void**>(&data_), workCap_ + 2 * inboxCap_));
  }
  if (trace) {
    GA_ERROR << "[phase r" << ctx_->rank << "] mesh:grow-exchange";
  }
  exchange();
  if (trace) {
    GA_ERROR << "[phase r" << ctx_->rank << "] mesh:grow-done";
  }
}

int XgmiMesh::allocFlags(int count) {
  GA_ENFORCE_LE(nextFlag_ + count, kNumFlags, "out of doorbell flags");
  int base = nextFlag_;
  nextFlag_ += count;
  return base;
}

void XgmiMesh::poisonFlags() {
  // Release OUR GPU-side waits AND every peer's: a rank that failed may
  // never write the flags its peers are spinning on, and their stuck
  // kernels would wedge device-wide syncs.
  std::vector<uint64_t> poison(kNumFlags, ~uint64_t(0) >> 1);
  (void)hipMemcpy(flags_, poison.data(), kNumFlags * 8,
                  hipMemcpyHostToDevice);
  for (size_t r = 0; r < peerFlags_.size(); r++) {
    if (static_cast<int>(r) != ctx_->rank && peerFlags_[r] != nullptr) {
      (void)hipMemcpy(peerFlags_[r], poison.data(), kNumFlags * 8,
                      hipMemcpyHostToDevice);
    }
  }
}

std::vector<uint64_t> XgmiMesh::readFlags(int n) {
  std::vector<uint64_t> out(n, 0);
  (void)hipMemcpy(out.data(), flags_, n * 8, hipMemcpyDeviceToHost);
  return out;
}

XgmiMesh::~XgmiMesh() {
  // No device-wide sync here: the owning engine's destructor drains its
  // streams before the mesh member is destroyed; a device-wide wait
  // could block on OTHER ranks' in-flight kernels when several ranks
  // share one GPU.
  releasePeers();
  std::lock_guard<std::mutex> lock(allocMutex());
  if (data_ != nullptr) {
    (void)hipFree(data_);
  }
  for (auto* p : retired_) {
    (void)hipFree(p);
  }
  if (flags_ != nullptr) {
    releaseFlagPage(flags_); // recycled, never freed (see acquireFlagPage)
  }
}

} // namespace hip
} // namespace glooamd
