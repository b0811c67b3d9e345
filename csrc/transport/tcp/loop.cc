#include "transport/tcp/loop.h"

#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <unistd.h>

#include <array>
#include <chrono>
#include <cstdlib>
#include <cstring>
#include <thread>

#include "common/logging.h"

namespace glooamd {
namespace tcp {

EpollLoop::EpollLoop() {
  if (const char* e = getenv("GLOO_AMD_BUSY_POLL_US")) {
    busyPollUs_ = atoi(e);
  }
  epfd_ = epoll_create1(EPOLL_CLOEXEC);
  GA_ENFORCE_GE(epfd_, 0, "epoll_create1: ", strerror(errno));
  evfd_ = eventfd(0, EFD_NONBLOCK | EFD_CLOEXEC);
  GA_ENFORCE_GE(evfd_, 0, "eventfd: ", strerror(errno));

  struct epoll_event ev;
  std::memset(&ev, 0, sizeof(ev));
  ev.events = EPOLLIN;
  ev.data.ptr = nullptr; // nullptr marks the wakeup fd
  GA_ENFORCE_EQ(epoll_ctl(epfd_, EPOLL_CTL_ADD, evfd_, &ev), 0);

  thread_ = std::thread([this] { run(); });
  threadId_ = thread_.get_id();
}

EpollLoop::~EpollLoop() {
  done_ = true;
  uint64_t one = 1;
  (void)!write(evfd_, &one, sizeof(one));
  if (thread_.joinable()) {
    thread_.join();
  }
  close(evfd_);
  close(epfd_);
}

void EpollLoop::registerDescriptor(int fd, uint32_t events, Handler* h) {
  struct epoll_event ev;
  std::memset(&ev, 0, sizeof(ev));
  ev.events = events;
  ev.data.ptr = h;
  int rv = epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev);
  GA_ENFORCE_EQ(rv, 0, "epoll_ctl ADD: ", strerror(errno));
}

void EpollLoop::modifyDescriptor(int fd, uint32_t events, Handler* h) {
  struct epoll_event ev;
  std::memset(&ev, 0, sizeof(ev));
  ev.events = events;
  ev.data.ptr = h;
  int rv = epoll_ctl(epfd_, EPOLL_CTL_MOD, fd, &ev);
  GA_ENFORCE_EQ(rv, 0, "epoll_ctl MOD: ", strerror(errno));
}

void EpollLoop::unregisterNoWait(int fd) {
  int rv = epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr);
  GA_ENFORCE_EQ(rv, 0, "epoll_ctl DEL: ", strerror(errno));
}

void EpollLoop::unregisterDescriptor(int fd) {
  int rv = epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr);
  GA_ENFORCE_EQ(rv, 0, "epoll_ctl DEL: ", strerror(errno));
  // If called off the loop thread, wait until the loop has moved past the
  // current dispatch batch so no handler for this fd can still be running.
  if (!inLoopThread()) {
    waitForTick();
  }
}

void EpollLoop::barrier() {
  if (!inLoopThread()) {
    waitForTick();
  }
}

void EpollLoop::waitForTick() {
  std::unique_lock<std::mutex> lock(mu_);
  uint64_t current = tick_;
  uint64_t one = 1;
  (void)!write(evfd_, &one, sizeof(one));
  cv_.wait(lock, [&] { return tick_ > current || done_.load(); });
}

void EpollLoop::defer(std::function<void()> fn) {
  {
    std::lock_guard<std::mutex> lock(mu_);
    deferred_.push_back(std::move(fn));
  }
  uint64_t one = 1;
  (void)!write(evfd_, &one, sizeof(one));
}

void EpollLoop::run() {
  std::array<struct epoll_event, 64> events;
  // After activity, spin with zero-timeout polls for a short window: the
  // next hop of an in-flight collective usually lands within microseconds,
  // and skipping the epoll sleep/wake saves ~5-15us per protocol hop.
  auto lastActivity = std::chrono::steady_clock::now();
  while (!done_.load()) {
    bool spinning = busyPollUs_ > 0 &&
        std::chrono::steady_clock::now() - lastActivity <
            std::chrono::microseconds(busyPollUs_);
    int n = epoll_wait(epfd_, events.data(), events.size(), spinning ? 0 : 100);
    if (n < 0 && errno != EINTR) {
      GA_ERROR << "epoll_wait: " << strerror(errno);
      break;
    }
    if (n == 0 && spinning) {
      // Cooperative spin: on an oversubscribed host (ranks >= cores,
      // e.g. threaded tests) a hard spin here starves the user threads
      // that produce the very work we are polling for. yield() costs
      // ~100ns when nothing else is runnable.
      std::this_thread::yield();
      continue; // nothing ready, no tick work owed to anyone
    }
    for (int i = 0; i < n; i++) {
      auto* h = static_cast<Handler*>(events[i].data.ptr);
      if (h == nullptr) {
        uint64_t val;
        (void)!read(evfd_, &val, sizeof(val));
        continue;
      }
      h->handleEvents(events[i].events);
    }
    if (n > 0) {
      lastActivity = std::chrono::steady_clock::now();
    }
    std::vector<std::function<void()>> fns;
    {
      std::lock_guard<std::mutex> lock(mu_);
      tick_++;
      fns.swap(deferred_);
    }
    cv_.notify_all();
    for (auto& fn : fns) {
      fn();
    }
  }
  // Drain cleanup work queued during shutdown (e.g. a Device dropping
  // pending unrouted connections) — skipping it would leak their fds.
  std::vector<std::function<void()>> fns;
  {
    std::lock_guard<std::mutex> lock(mu_);
    fns.swap(deferred_);
  }
  for (auto& fn : fns) {
    fn();
  }
  // final tick so waiters don't hang at shutdown
  {
    std::lock_guard<std::mutex> lock(mu_);
    tick_++;
  }
  cv_.notify_all();
}

} // namespace tcp
} // namespace glooamd
