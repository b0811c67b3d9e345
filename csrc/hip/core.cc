#include "hip/core.h"

#include <map>
#include <memory>
#include <tuple>

namespace glooamd {
namespace hip {

std::mutex& allocMutex() {
  static std::mutex mu;
  return mu;
}

bool available() {
  int n = 0;
  return hipGetDeviceCount(&n) == hipSuccess && n > 0;
}

int deviceCount() {
  int n = 0;
  GA_HIP_CHECK(hipGetDeviceCount(&n));
  return n;
}

std::string gpuPCIBusID(int device) {
  char buf[64] = {0};
  GA_HIP_CHECK(hipDeviceGetPCIBusId(buf, sizeof(buf), device));
  return std::string(buf);
}

HipStream::HipStream(int device, bool highPriority) : device_(device) {
  if (device_ >= 0) {
    GA_HIP_CHECK(hipSetDevice(device_));
  } else {
    GA_HIP_CHECK(hipGetDevice(&device_));
  }
  if (highPriority) {
    int least = 0, greatest = 0;
    GA_HIP_CHECK(hipDeviceGetStreamPriorityRange(&least, &greatest));
    GA_HIP_CHECK(hipStreamCreateWithPriority(
        &stream_, hipStreamNonBlocking, greatest));
  } else {
    GA_HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
  }
  GA_HIP_CHECK(hipEventCreateWithFlags(&event_, hipEventDisableTiming));
}

HipStream::HipStream(HipStream&& o) noexcept
    : device_(o.device_), stream_(o.stream_), event_(o.event_) {
  o.stream_ = nullptr;
  o.event_ = nullptr;
}

HipStream::~HipStream() {
  if (stream_ != nullptr) {
    (void)hipStreamSynchronize(stream_);
    (void)hipStreamDestroy(stream_);
  }
  if (event_ != nullptr) {
    (void)hipEventDestroy(event_);
  }
}

void HipStream::synchronize() {
  GA_HIP_CHECK(hipStreamSynchronize(stream_));
}

void HipStream::recordAndWait(HipStream& other) {
  GA_HIP_CHECK(hipEventRecord(event_, stream_));
  GA_HIP_CHECK(hipStreamWaitEvent(other.stream_, event_, 0));
}

HipStream* pooledStream(const void* key, int device, int idx) {
  GA_ENFORCE_GE(idx, 0);
  GA_ENFORCE_LT(idx, kStreamPoolSize, "stream pool exhausted");
  static std::mutex mu;
  // Intentionally leaked: destroying streams during static teardown (after
  // the HIP runtime unloads) is unsafe.
  static auto* pool = new std::map<std::tuple<const void*, int, int>,
                                   std::unique_ptr<HipStream>>();
  std::lock_guard<std::mutex> lock(mu);
  auto k = std::make_tuple(key, device, idx);
  auto it = pool->find(k);
  if (it == pool->end()) {
    it = pool->emplace(k, std::make_unique<HipStream>(
                              device, /*highPriority=*/idx == 0))
             .first;
  }
  return it->second.get();
}

HipEvent::HipEvent(int device) {
  if (device >= 0) {
    GA_HIP_CHECK(hipSetDevice(device));
  }
  GA_HIP_CHECK(hipEventCreateWithFlags(&event_, hipEventDisableTiming));
}

HipEvent::~HipEvent() {
  if (event_ != nullptr) {
    (void)hipEventDestroy(event_);
  }
}

void HipEvent::record(hipStream_t s) {
  GA_HIP_CHECK(hipEventRecord(event_, s));
}

void HipEvent::streamWait(hipStream_t s) {
  GA_HIP_CHECK(hipStreamWaitEvent(s, event_, 0));
}

bool HipEvent::query() {
  hipError_t e = hipEventQuery(event_);
  if (e == hipSuccess) {
    return true;
  }
  if (e == hipErrorNotReady) {
    return false;
  }
  GA_HIP_CHECK(e);
  return false;
}

void HipEvent::synchronize() {
  GA_HIP_CHECK(hipEventSynchronize(event_));
}

} // namespace hip
} // namespace glooamd
