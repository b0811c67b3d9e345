// HIP host-side primitives: error checking, streams, events, typed
// device/pinned-host pointers.
//
// Capability parity with the reference CUDA layer (gloo/cuda.h:56-245:
// CudaStream / CudaDevicePointer / CudaHostPointer, gloo/cuda.h:40-54
// CudaShared allocation mutex). Differences by design: streams are
// created non-blocking without the ROCm-hang priority workaround the
// reference needed (gloo/cuda.cu:46-55), and there is no NCCL delegate.
#pragma once

#include <hip/hip_runtime.h>

#include <cstddef>
#include <mutex>

#include "common/logging.h"

#define GA_HIP_CHECK(expr)                                          \
  do {                                                              \
    hipError_t ga_err_ = (expr);                                    \
    if (ga_err_ != hipSuccess) {                                    \
      GA_THROW_IO(                                                  \
          "HIP error ",                                             \
          hipGetErrorName(ga_err_),                                 \
          " (",                                                     \
          static_cast<int>(ga_err_),                                \
          "): ",                                                    \
          hipGetErrorString(ga_err_),                               \
          " at " #expr);                                            \
    }                                                               \
  } while (0)

namespace glooamd {
namespace hip {

// Serializes hipMalloc/hipFree against in-flight collectives (reference
// parity: CudaShared mutex, gloo/docs/cuda.md).
std::mutex& allocMutex();

bool available(); // any HIP device present (never throws)
int deviceCount();
// PCI bus id of a GPU ("0000:03:00.0"), for NIC-locality decisions
// (reference cuda_private.cu:96-114 getCudaPCIBusID parity).
std::string gpuPCIBusID(int device);

class HipStream {
 public:
  explicit HipStream(int device = -1, bool highPriority = false);
  ~HipStream();
  HipStream(const HipStream&) = delete;
  HipStream(HipStream&& o) noexcept;

  hipStream_t stream() const {
    return stream_;
  }
  int device() const {
    return device_;
  }
  void synchronize();
  // Record this stream's completion event and make `other` wait on it.
  void recordAndWait(HipStream& other);

 private:
  int device_{-1};
  hipStream_t stream_{nullptr};
  hipEvent_t event_{nullptr};
};

// Process-wide stream pool. ROCm maps HIP streams onto at most
// GPU_MAX_HW_QUEUES (default 4, we request 8) hardware queues per
// process+device; streams beyond that SHARE a queue, and a doorbell
// spin kernel then head-of-line-blocks whatever shares its queue —
// instant deadlock for flag-wait protocols. Capping each communicator
// at 8 pooled streams (index 0 = high priority) keeps every stream on
// its own hardware queue. Pool key: one per (communicator, device) —
// different algorithm objects of one communicator intentionally reuse
// the same streams (collectives on a communicator are serialized).
constexpr int kStreamPoolSize = 8;
HipStream* pooledStream(const void* key, int device, int idx);
// Return a destroyed context's pooled streams to the per-(device,idx)
// freelist (drained first). Called from glooamd::Context's destructor
// so the process-lifetime stream count stays at the live working set.
void releasePooledStreams(const void* key);

class HipEvent {
 public:
  explicit HipEvent(int device);
  ~HipEvent();
  HipEvent(const HipEvent&) = delete;
  void record(hipStream_t s);
  void streamWait(hipStream_t s);
  bool query();
  void synchronize();
  hipEvent_t event() const {
    return event_;
  }

 private:
  hipEvent_t event_{nullptr};
};

// Owned device allocation (or borrowed view).
template <typename T>
class HipDevicePointer {
 public:
  static HipDevicePointer alloc(size_t count) {
    void* p = nullptr;
    {
      std::lock_guard<std::mutex> lock(allocMutex());
      GA_HIP_CHECK(hipMalloc(&p, count * sizeof(T)));
    }
    return HipDevicePointer(static_cast<T*>(p), count, true);
  }
  static HipDevicePointer view(T* ptr, size_t count) {
    return HipDevicePointer(ptr, count, false);
  }
  HipDevicePointer(HipDevicePointer&& o) noexcept
      : ptr_(o.ptr_), count_(o.count_), owned_(o.owned_) {
    o.ptr_ = nullptr;
    o.owned_ = false;
  }
  ~HipDevicePointer() {
    if (owned_ && ptr_ != nullptr) {
      std::lock_guard<std::mutex> lock(allocMutex());
      (void)hipFree(ptr_);
    }
  }
  T* operator*() const {
    return ptr_;
  }
  T* get() const {
    return ptr_;
  }
  size_t getCount() const {
    return count_;
  }
  HipDevicePointer range(size_t offset, size_t count) const {
    GA_ENFORCE_LE(offset + count, count_);
    return HipDevicePointer(ptr_ + offset, count, false);
  }

 private:
  HipDevicePointer(T* ptr, size_t count, bool owned)
      : ptr_(ptr), count_(count), owned_(owned) {}
  T* ptr_;
  size_t count_;
  bool owned_;
};

// Pinned host allocation.
template <typename T>
class HipHostPointer {
 public:
  static HipHostPointer alloc(size_t count) {
    void* p = nullptr;
    GA_HIP_CHECK(hipHostMalloc(&p, count * sizeof(T)));
    return HipHostPointer(static_cast<T*>(p), count, true);
  }
  HipHostPointer(HipHostPointer&& o) noexcept
      : ptr_(o.ptr_), count_(o.count_), owned_(o.owned_) {
    o.ptr_ = nullptr;
    o.owned_ = false;
  }
  ~HipHostPointer() {
    if (owned_ && ptr_ != nullptr) {
      (void)hipHostFree(ptr_);
    }
  }
  T* operator*() const {
    return ptr_;
  }
  T* get() const {
    return ptr_;
  }
  size_t getCount() const {
    return count_;
  }

 private:
  HipHostPointer(T* ptr, size_t count, bool owned)
      : ptr_(ptr), count_(count), owned_(owned) {}
  T* ptr_;
  size_t count_;
  bool owned_;
};

} // namespace hip
} // namespace glooamd
