#include "transport/tcp/uv_loop.h"

#include <sys/epoll.h>
#include <uv.h>

#include <cstdlib>

#include <atomic>
#include <condition_variable>
#include <mutex>
#include <thread>
#include <unordered_map>
#include <vector>

#include "common/logging.h"

namespace glooamd {
namespace tcp {

namespace {

class UvLoop : public Loop {
 public:
  UvLoop() {
    if (const char* e = getenv("GLOO_AMD_BUSY_POLL_US")) {
      busyPollUs_ = atoi(e);
    }
    GA_ENFORCE_EQ(uv_loop_init(&loop_), 0);
    loop_.data = this;
    // Busy-poll window: while the idle handle is active uv_run polls
    // with zero timeout; activity (re)arms it, the idle callback
    // disarms once the window has passed (EpollLoop parity).
    idle_.data = this;
    GA_ENFORCE_EQ(uv_idle_init(&loop_, &idle_), 0);
    async_.data = this;
    GA_ENFORCE_EQ(
        uv_async_init(&loop_, &async_,
                      [](uv_async_t* a) {
                        static_cast<UvLoop*>(a->data)->onWake();
                      }),
        0);
    // Tick after every iteration so off-thread unregister can wait for
    // in-flight handlers to finish (EpollLoop::waitForTick parity).
    prepare_.data = this;
    GA_ENFORCE_EQ(uv_prepare_init(&loop_, &prepare_), 0);
    GA_ENFORCE_EQ(
        uv_prepare_start(&prepare_,
                         [](uv_prepare_t* p) {
                           static_cast<UvLoop*>(p->data)->onTick();
                         }),
        0);
    thread_ = std::thread([this] { run(); });
    threadId_ = thread_.get_id();
  }

  ~UvLoop() override {
    defer([this] {
      for (auto& kv : polls_) {
        uv_poll_stop(kv.second);
        uv_close(reinterpret_cast<uv_handle_t*>(kv.second),
                 [](uv_handle_t* h) {
                   delete reinterpret_cast<uv_poll_t*>(h);
                 });
      }
      polls_.clear();
      if (idleActive_) {
        uv_idle_stop(&idle_);
        idleActive_ = false;
      }
      uv_close(reinterpret_cast<uv_handle_t*>(&idle_), nullptr);
      uv_prepare_stop(&prepare_);
      uv_close(reinterpret_cast<uv_handle_t*>(&prepare_), nullptr);
      uv_close(reinterpret_cast<uv_handle_t*>(&async_), nullptr);
      done_ = true;
    });
    if (thread_.joinable()) {
      thread_.join();
    }
    uv_loop_close(&loop_);
  }

  void registerDescriptor(int fd, uint32_t events, Handler* h) override {
    runOnLoop([this, fd, events, h] {
      auto* poll = new uv_poll_t;
      poll->data = h;
      GA_ENFORCE_EQ(uv_poll_init(&loop_, poll, fd), 0, "uv_poll_init");
      polls_[fd] = poll;
      startPoll(poll, events);
    });
  }

  void modifyDescriptor(int fd, uint32_t events, Handler* h) override {
    runOnLoop([this, fd, events, h] {
      auto it = polls_.find(fd);
      if (it == polls_.end()) {
        return;
      }
      it->second->data = h;
      startPoll(it->second, events);
    });
  }

  void unregisterDescriptor(int fd) override {
    if (inLoopThread()) {
      removePoll(fd);
    } else {
      runOnLoopAndWait([this, fd] { removePoll(fd); });
    }
  }

  void unregisterNoWait(int fd) override {
    defer([this, fd] { removePoll(fd); });
  }

  void barrier() override {
    if (!inLoopThread()) {
      // Deferred fns run FIFO on the loop thread, serialized with poll
      // callbacks: when this no-op has run, every earlier removePoll has
      // too and no handler dispatch is in flight.
      runOnLoopAndWait([] {});
    }
  }

  void defer(std::function<void()> fn) override {
    {
      std::lock_guard<std::mutex> lock(mu_);
      deferred_.push_back(std::move(fn));
    }
    uv_async_send(&async_);
  }

  bool inLoopThread() const override {
    return std::this_thread::get_id() == threadId_;
  }

 private:
  void startPoll(uv_poll_t* poll, uint32_t events) {
    int mask = 0;
    if (events & EPOLLIN) {
      mask |= UV_READABLE;
    }
    if (events & EPOLLOUT) {
      mask |= UV_WRITABLE;
    }
    uv_poll_start(poll, mask, [](uv_poll_t* p, int status, int uvEvents) {
      auto* h = static_cast<Handler*>(p->data);
      uint32_t ev = 0;
      if (status < 0) {
        ev |= EPOLLERR;
      } else {
        if (uvEvents & UV_READABLE) {
          ev |= EPOLLIN;
        }
        if (uvEvents & UV_WRITABLE) {
          ev |= EPOLLOUT;
        }
      }
      if (ev != 0 && h != nullptr) {
        static_cast<UvLoop*>(p->loop->data)->noteActivity();
        h->handleEvents(ev);
      }
    });
  }

  // loop thread only
  void noteActivity() {
    lastActivityNs_ = uv_hrtime();
    if (busyPollUs_ > 0 && !idleActive_) {
      idleActive_ = true;
      uv_idle_start(&idle_, [](uv_idle_t* i) {
        auto* self = static_cast<UvLoop*>(i->data);
        if (uv_hrtime() - self->lastActivityNs_ >
            static_cast<uint64_t>(self->busyPollUs_) * 1000) {
          uv_idle_stop(i);
          self->idleActive_ = false;
        }
      });
    }
  }

  void removePoll(int fd) {
    auto it = polls_.find(fd);
    if (it == polls_.end()) {
      return;
    }
    uv_poll_stop(it->second);
    it->second->data = nullptr;
    uv_close(reinterpret_cast<uv_handle_t*>(it->second), [](uv_handle_t* h) {
      delete reinterpret_cast<uv_poll_t*>(h);
    });
    polls_.erase(it);
  }

  void runOnLoop(std::function<void()> fn) {
    if (inLoopThread()) {
      fn();
    } else {
      defer(std::move(fn));
    }
  }

  void runOnLoopAndWait(std::function<void()> fn) {
    std::mutex m;
    std::condition_variable cv;
    bool doneFlag = false;
    defer([&] {
      fn();
      // Notify while holding the mutex: the waiter re-acquires it before
      // returning, so the stack cv cannot be destroyed mid-broadcast.
      std::lock_guard<std::mutex> lock(m);
      doneFlag = true;
      cv.notify_all();
    });
    std::unique_lock<std::mutex> lock(m);
    cv.wait(lock, [&] { return doneFlag; });
  }

  void onWake() {
    std::vector<std::function<void()>> fns;
    {
      std::lock_guard<std::mutex> lock(mu_);
      fns.swap(deferred_);
    }
    for (auto& fn : fns) {
      fn();
    }
    if (done_) {
      uv_stop(&loop_);
    }
  }

  void onTick() {
    {
      std::lock_guard<std::mutex> lock(mu_);
      tick_++;
    }
    tickCv_.notify_all();
  }

  void run() {
    uv_run(&loop_, UV_RUN_DEFAULT);
    // Drain handle closes queued by the destructor.
    uv_run(&loop_, UV_RUN_NOWAIT);
    uv_run(&loop_, UV_RUN_NOWAIT);
  }

  uv_loop_t loop_;
  uv_async_t async_;
  uv_prepare_t prepare_;
  uv_idle_t idle_;
  bool idleActive_{false}; // loop thread only
  uint64_t lastActivityNs_{0};
  int busyPollUs_{200};
  std::thread thread_;
  std::thread::id threadId_;
  std::atomic<bool> done_{false};
  std::mutex mu_;
  std::condition_variable tickCv_;
  uint64_t tick_{0};
  std::vector<std::function<void()>> deferred_;
  std::unordered_map<int, uv_poll_t*> polls_; // loop thread only
};

} // namespace

std::unique_ptr<Loop> makeUvLoop() {
  return std::make_unique<UvLoop>();
}

} // namespace tcp
} // namespace glooamd
