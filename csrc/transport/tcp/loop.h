// Epoll event loop: one thread per tcp Device.
// Capability parity with reference gloo/transport/tcp/loop.cc:103-220
// (epoll_wait + handler dispatch + deferred-function wakeup + safe
// off-thread unregister). Re-designed: eventfd wakeup instead of a pipe,
// tick-counter barrier for teardown safety.
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
  Loop();
  ~Loop();

  // Register fd with EPOLLIN/EPOLLOUT mask; handler invoked on loop thread.
  void registerDescriptor(int fd, uint32_t events, Handler* h);
  void modifyDescriptor(int fd, uint32_t events, Handler* h);

  // After return, the handler for fd is guaranteed not running and will
  // never run again (blocks one dispatch tick when called off-thread).
  void unregisterDescriptor(int fd);

  // epoll DEL only — no tick wait. Safe to call while holding locks the
  // loop thread may contend on; pair with defer() for the close().
  void unregisterNoWait(int fd);

  // Run fn on the loop thread soon.
  void defer(std::function<void()> fn);

  bool inLoopThread() const {
    return std::this_thread::get_id() == threadId_;
  }

 private:
  void run();
  void waitForTick();

  int epfd_{-1};
  int evfd_{-1};
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
