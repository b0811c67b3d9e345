#include "hip/core.h"

#include <chrono>
#include <map>
#include <thread>
#include <vector>
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
    // Bounded drain: an unbounded sync on a wedged stream would hang
    // teardown forever; hipStreamDestroy itself defers resource release
    // until the remaining work (if any) completes.
    const auto dl =
        std::chrono::steady_clock::now() + std::chrono::seconds(2);
    while (hipStreamQuery(stream_) == hipErrorNotReady &&
           std::chrono::steady_clock::now() < dl) {
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
    (void)hipGetLastError();
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

namespace {
std::mutex& poolMutex() {
  static std::mutex mu;
  return mu;
}
// Intentionally leaked: destroying streams during static teardown (after
// the HIP runtime unloads) is unsafe.
std::map<std::tuple<const void*, int, int>, std::unique_ptr<HipStream>>&
streamPool() {
  static auto* pool = new std::map<std::tuple<const void*, int, int>,
                                   std::unique_ptr<HipStream>>();
  return *pool;
}
// Streams of destroyed contexts, reusable per (device, idx). Bounding
// the process-lifetime stream count matters: HIP multiplexes streams
// onto a small number of hardware queues, and once two ACTIVE streams
// share a queue, a cross-stream event-wait packet at one queue's head
// can block the very stream that would satisfy it (observed as flaky
// deadlocks in engine-recreation loops once stream count grew past the
// queue count). Reuse keeps the total at the live working set.
std::map<std::pair<int, int>, std::vector<std::unique_ptr<HipStream>>>&
streamFreelist() {
  static auto* fl = new std::map<std::pair<int, int>,
                                 std::vector<std::unique_ptr<HipStream>>>();
  return *fl;
}
} // namespace

HipStream* pooledStream(const void* key, int device, int idx) {
  GA_ENFORCE_GE(idx, 0);
  GA_ENFORCE_LT(idx, kStreamPoolSize, "stream pool exhausted");
  std::lock_guard<std::mutex> lock(poolMutex());
  auto k = std::make_tuple(key, device, idx);
  auto& pool = streamPool();
  auto it = pool.find(k);
  if (it == pool.end()) {
    auto& fl = streamFreelist()[{device, idx}];
    if (!fl.empty()) {
      it = pool.emplace(k, std::move(fl.back())).first;
      fl.pop_back();
    } else {
      it = pool.emplace(k, std::make_unique<HipStream>(
                               device, /*highPriority=*/idx == 0))
               .first;
    }
  }
  return it->second.get();
}

void releasePooledStreams(const void* key) {
  // Streams that still have work (a poisoned run's tail, or a wedged
  // queue) are QUARANTINED, not reused and not destroyed: syncing or
  // destroying them could hang this (fail-fast) teardown path forever.
  static auto* graveyard = new std::vector<std::unique_ptr<HipStream>>();
  std::lock_guard<std::mutex> lock(poolMutex());
  auto& pool = streamPool();
  for (auto it = pool.begin(); it != pool.end();) {
    if (std::get<0>(it->first) == key) {
      if (hipStreamQuery(it->second->stream()) == hipSuccess) {
        streamFreelist()[{std::get<1>(it->first), std::get<2>(it->first)}]
            .push_back(std::move(it->second));
      } else {
        (void)hipGetLastError(); // swallow hipErrorNotReady
        graveyard->push_back(std::move(it->second));
      }
      it = pool.erase(it);
    } else {
      ++it;
    }
  }
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
