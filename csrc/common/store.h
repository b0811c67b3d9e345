// Abstract key/value store used for rendezvous.
// Capability parity with reference gloo/common/store.h:20-53 and
// gloo/rendezvous/store.h:25-74 (set/get/wait + batched multi ops).
#pragma once

#include <chrono>
#include <stdexcept>
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

  // --- v2 store API parity (reference gloo/rendezvous/store.h:25-74) ---
  // Stores that override append/add atomically report true here.
  virtual bool hasV2() const {
    return false;
  }
  // Atomically concatenate data to the key's value (creates it if absent).
  virtual void append(const std::string& key, const std::vector<char>& data) {
    (void)key;
    (void)data;
    throw std::logic_error("store does not support append (no v2 API)");
  }
  // Atomically add delta to an integer-valued key; returns the new value.
  virtual int64_t add(const std::string& key, int64_t delta) {
    (void)key;
    (void)delta;
    throw std::logic_error("store does not support add (no v2 API)");
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
