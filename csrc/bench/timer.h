// Latency sampling + percentile distribution for the benchmark tool.
// Capability parity with reference gloo/benchmark/timer.h:21-110.
#pragma once

#include <algorithm>
#include <chrono>
#include <cstdint>
#include <vector>

namespace glooamd {
namespace bench {

class Timer {
 public:
  void start() {
    t0_ = std::chrono::steady_clock::now();
  }
  int64_t lapNs() const {
    return std::chrono::duration_cast<std::chrono::nanoseconds>(
               std::chrono::steady_clock::now() - t0_)
        .count();
  }

 private:
  std::chrono::steady_clock::time_point t0_;
};

class Samples {
 public:
  static constexpr size_t kMax = 100000;

  void add(int64_t ns) {
    if (ns_.size() < kMax) {
      ns_.push_back(ns);
    }
  }
  size_t size() const {
    return ns_.size();
  }
  void clear() {
    ns_.clear();
  }
  const std::vector<int64_t>& raw() const {
    return ns_;
  }

 private:
  std::vector<int64_t> ns_;
};

struct Distribution {
  explicit Distribution(const Samples& s) : sorted(s.raw()) {
    std::sort(sorted.begin(), sorted.end());
  }
  size_t size() const {
    return sorted.size();
  }
  int64_t min() const {
    return sorted.empty() ? 0 : sorted.front();
  }
  int64_t max() const {
    return sorted.empty() ? 0 : sorted.back();
  }
  int64_t percentile(double p) const {
    if (sorted.empty()) {
      return 0;
    }
    size_t idx = static_cast<size_t>(p * sorted.size());
    return sorted[std::min(idx, sorted.size() - 1)];
  }
  int64_t sum() const {
    int64_t t = 0;
    for (auto v : sorted) {
      t += v;
    }
    return t;
  }
  std::vector<int64_t> sorted;
};

} // namespace bench
} // namespace glooamd
