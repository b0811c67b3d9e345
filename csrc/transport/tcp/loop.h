// Event loops driving the tcp transport.
// Capability parity with reference gloo/transport/tcp/loop.cc:103-220
// (epoll) and gloo/transport/uv/ (libuv): the SAME pair protocol runs
// over either loop; `Loop` is the interface, `EpollLoop` the primary
// Linux implementation, `UvLoop` (csrc/transport/tcp/uv_loop.*) the
// libuv-driven one.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

namespace glooamd {
namespace tcp {

class Handler {
 public:
  virtual ~Handler() = default;
  virtual void handleEvents(uint32_t events) = 0;
};

class Loop {
 public:
  virtual ~Loop() = default;

  // Register fd with EPOLLIN/EPOLLOUT mask; handler invoked on loop thread.
  virtual void registerDescriptor(int fd, uint32_t events, Handler* h) = 0;
  virtual void modifyDescriptor(int fd, uint32_t events, Handler* h) = 0;

  // After return, the handler for fd is guaranteed not running and will
  // never run again (may block one dispatch tick when called off-thread).
  virtual void unregisterDescriptor(int fd) = 0;

  // Detach only — no tick wait. Safe while holding locks the loop thread
  // may contend on; pair with defer() for the close().
  virtual void unregisterNoWait(int fd) = 0;

  // Run fn on the loop thread soon.
  virtual void defer(std::function<void()> fn) = 0;

  // Returns once the loop has moved past its current dispatch batch (and,
  // for deferred-removal loops, past previously deferred teardown), so a
  // handler unregistered via unregisterNoWait can be safely deleted.
  // No-op from the loop thread itself.
  virtual void barrier() = 0;

  virtual bool inLoopThread() const = 0;
};

class EpollLoop : public Loop {
 public:
  EpollLoop();
  ~EpollLoop() override;

  void registerDescriptor(int fd, uint32_t events, Handler* h) override;
  void modifyDescriptor(int fd, uint32_t events, Handler* h) override;
  void unregisterDescriptor(int fd) override;
  void unregisterNoWait(int fd) override;
  void defer(std::function<void()> fn) override;
  void barrier() override;

  bool inLoopThread() const override {
    return std::this_thread::get_id() == threadId_;
  }

 private:
  void run();
  void waitForTick();

  int epfd_{-1};
  int evfd_{-1};
  int busyPollUs_{200}; // GLOO_AMD_BUSY_POLL_US; 0 disables
  std::atomic<bool> done_{false};
  std::thread thread_;
  std::thread::id threadId_;

  std::mutex mu_;
  std::condition_variable cv_;
  uint64_t tick_{0};
  std::vector<std::function<void()>> deferred_;
};

} // namespace tcp
} // namespace glooamd
