// Rendezvous stores: HashStore (in-process), FileStore (shared fs),
// PrefixStore (namespacing), TcpStore (client/server KV over TCP).
//
// Capability parity with reference gloo/rendezvous/{hash,file,prefix}_store
// and RedisStore. The network store is re-designed as a self-contained
// TcpStore (server thread + tiny length-prefixed protocol) instead of a
// Redis client: a single 8xMI355X node or a torchrun launch has no Redis,
// but always has the master address/port pair.
#pragma once

#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <thread>
#include <unordered_map>

#include "common/store.h"

namespace glooamd {

class HashStore : public IStore {
 public:
  void set(const std::string& key, const std::vector<char>& data) override;
  std::vector<char> get(const std::string& key) override;
  void wait(
      const std::vector<std::string>& keys,
      const std::chrono::milliseconds& timeout) override;
  bool hasV2() const override {
    return true;
  }
  void append(const std::string& key, const std::vector<char>& data) override;
  int64_t add(const std::string& key, int64_t delta) override;

 private:
  std::mutex mu_;
  std::condition_variable cv_;
  std::unordered_map<std::string, std::vector<char>> map_;
};

class FileStore : public IStore {
 public:
  explicit FileStore(const std::string& path);
  void set(const std::string& key, const std::vector<char>& data) override;
  std::vector<char> get(const std::string& key) override;
  void wait(
      const std::vector<std::string>& keys,
      const std::chrono::milliseconds& timeout) override;
  bool hasV2() const override {
    return true;
  }
  // Atomic across processes via flock on a per-store lock file.
  void append(const std::string& key, const std::vector<char>& data) override;
  int64_t add(const std::string& key, int64_t delta) override;

 private:
  std::string objectPath(const std::string& key) const;
  bool check(const std::vector<std::string>& keys) const;
  std::string basePath_;
};

class PrefixStore : public IStore {
 public:
  PrefixStore(const std::string& prefix, std::shared_ptr<IStore> store);
  void set(const std::string& key, const std::vector<char>& data) override;
  std::vector<char> get(const std::string& key) override;
  void wait(
      const std::vector<std::string>& keys,
      const std::chrono::milliseconds& timeout) override;
  bool hasV2() const override {
    return store_->hasV2();
  }
  void append(const std::string& key, const std::vector<char>& data) override;
  int64_t add(const std::string& key, int64_t delta) override;

 private:
  std::string prefix_;
  std::shared_ptr<IStore> store_;
};

// TCP key/value store. One process runs the server (isServer=true, usually
// rank 0); every process (including the server's own) connects as a client.
// Wire format: u8 op ('S','G','W','C','A','I') + u32 keylen + key
// [+ u32 vallen + val | i64 delta]; GET blocks server-side until the
// key exists; 'C' checks without blocking; 'A' appends, 'I' atomically adds and returns the value.
class TcpStore : public IStore {
 public:
  TcpStore(
      const std::string& host,
      int port,
      bool isServer,
      std::chrono::milliseconds timeout = std::chrono::milliseconds(60000));
  ~TcpStore() override;

  void set(const std::string& key, const std::vector<char>& data) override;
  std::vector<char> get(const std::string& key) override;
  void wait(
      const std::vector<std::string>& keys,
      const std::chrono::milliseconds& timeout) override;
  bool hasV2() const override {
    return true;
  }
  void append(const std::string& key, const std::vector<char>& data) override;
  int64_t add(const std::string& key, int64_t delta) override;

 private:
  class Server;
  std::unique_ptr<Server> server_;
  int clientFd_{-1};
  std::mutex clientMu_;
  std::chrono::milliseconds timeout_;
};

} // namespace glooamd
