// Abstract key/value store used for rendezvous.
// Capability parity with reference gloo/common/store.h:20-53 and
// gloo/rendezvous/store.h:25-74 (set/get/wait + batched multi ops).
#pragma once

#include <chrono>
#include <string>
#include <vector>

namespace glooamd {

class IStore {
 public:
  static constexpr std::chrono::milliseconds kDefaultTimeout{30000};

  virtual ~IStore() = default;

  virtual void set(const std::string& key, const std::vector<char>& data) = 0;

  // Blocks until the key exists (up to timeout), then returns its value.
  virtual std::vector<char> get(const std::string& key) = 0;

  // Blocks until all keys exist. Throws TimeoutException on expiry.
  virtual void wait(
      const std::vector<std::string>& keys,
      const std::chrono::milliseconds& timeout) = 0;
  void wait(const std::vector<std::string>& keys) {
    wait(keys, kDefaultTimeout);
  }

  // Batched variants (v2 store API parity); default to loops.
  virtual void multiSet(
      const std::vector<std::string>& keys,
      const std::vector<std::vector<char>>& values) {
    for (size_t i = 0; i < keys.size(); i++) {
      set(keys[i], values[i]);
    }
  }
  virtual std::vector<std::vector<char>> multiGet(
      const std::vector<std::string>& keys) {
    std::vector<std::vector<char>> out;
    out.reserve(keys.size());
    for (const auto& k : keys) {
      out.push_back(get(k));
    }
    return out;
  }
};

} // namespace glooamd
