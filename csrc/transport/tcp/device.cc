#include "transport/tcp/device.h"

#include <fcntl.h>
#include <netdb.h>
#include <sys/un.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <cstring>

#include "common/logging.h"
#include "transport/tcp/context.h"
#include "transport/tcp/uv_loop.h"

namespace glooamd {
namespace tcp {

void setNonBlocking(int fd) {
  int flags = fcntl(fd, F_GETFL, 0);
  GA_ENFORCE_GE(flags, 0, "fcntl(F_GETFL): ", strerror(errno));
  GA_ENFORCE_GE(
      fcntl(fd, F_SETFL, flags | O_NONBLOCK), 0, "fcntl: ", strerror(errno));
}

void setSocketOptions(int fd) {
  int on = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &on, sizeof(on));
  // Large kernel buffers: localhost/loopback collective payloads are
  // multi-MB; autotuning handles the rest.
  int bufsize = 16 << 20;
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &bufsize, sizeof(bufsize));
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &bufsize, sizeof(bufsize));
}

std::shared_ptr<TcpDevice> createTcpDevice(const TcpAttr& attr) {
  return std::make_shared<TcpDevice>(attr);
}

TcpDevice::TcpDevice(const TcpAttr& attr) {
  if (attr.useLibuv) {
    loop_ = makeUvLoop();
  } else {
    loop_ = std::make_unique<EpollLoop>();
  }
  if (attr.useUds) {
    // Abstract-namespace unix socket: unique zero-padded name, no
    // filesystem residue. bind/connect both use sizeof(sockaddr_un) so
    // the (padded) name length is consistent end to end.
    listenFd_ = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
    GA_ENFORCE_GE(listenFd_, 0, "socket(AF_UNIX): ", strerror(errno));
    struct sockaddr_un sun;
    std::memset(&sun, 0, sizeof(sun));
    sun.sun_family = AF_UNIX;
    static std::atomic<uint64_t> ctr{0};
    snprintf(
        sun.sun_path + 1,
        sizeof(sun.sun_path) - 1,
        "gloo_amd_%d_%llu",
        static_cast<int>(getpid()),
        static_cast<unsigned long long>(ctr.fetch_add(1)));
    int rv = bind(
        listenFd_, reinterpret_cast<struct sockaddr*>(&sun), sizeof(sun));
    GA_ENFORCE_EQ(rv, 0, "bind(uds): ", strerror(errno));
    rv = listen(listenFd_, 1024);
    GA_ENFORCE_EQ(rv, 0, "listen(uds): ", strerror(errno));
    std::memset(&advertised_, 0, sizeof(advertised_));
    std::memcpy(&advertised_, &sun, sizeof(sun));
    setNonBlocking(listenFd_);
    loop_->registerDescriptor(listenFd_, EPOLLIN, this);
    return;
  }
  std::string host = attr.hostname.empty() ? "127.0.0.1" : attr.hostname;

  struct addrinfo hints;
  std::memset(&hints, 0, sizeof(hints));
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  int rv = getaddrinfo(host.c_str(), nullptr, &hints, &res);
  GA_ENFORCE_EQ(rv, 0, "getaddrinfo(", host, "): ", gai_strerror(rv));

  listenFd_ = socket(res->ai_family, SOCK_STREAM | SOCK_CLOEXEC, 0);
  GA_ENFORCE_GE(listenFd_, 0, "socket: ", strerror(errno));
  int on = 1;
  setsockopt(listenFd_, SOL_SOCKET, SO_REUSEADDR, &on, sizeof(on));

  // Bind the advertised host (loopback by default) with the requested port.
  struct sockaddr_storage bindAddr;
  std::memset(&bindAddr, 0, sizeof(bindAddr));
  std::memcpy(&bindAddr, res->ai_addr, res->ai_addrlen);
  socklen_t addrlen = res->ai_addrlen;
  if (bindAddr.ss_family == AF_INET) {
    reinterpret_cast<struct sockaddr_in*>(&bindAddr)->sin_port =
        htons(attr.port);
  } else if (bindAddr.ss_family == AF_INET6) {
    reinterpret_cast<struct sockaddr_in6*>(&bindAddr)->sin6_port =
        htons(attr.port);
  }
  freeaddrinfo(res);

  rv = bind(listenFd_, reinterpret_cast<struct sockaddr*>(&bindAddr), addrlen);
  GA_ENFORCE_EQ(rv, 0, "bind(", host, "): ", strerror(errno));
  rv = listen(listenFd_, 1024);
  GA_ENFORCE_EQ(rv, 0, "listen: ", strerror(errno));

  // Advertised address: the bound sockaddr (includes the assigned port).
  addrlen = sizeof(advertised_);
  std::memset(&advertised_, 0, sizeof(advertised_));
  rv = getsockname(
      listenFd_, reinterpret_cast<struct sockaddr*>(&advertised_), &addrlen);
  GA_ENFORCE_EQ(rv, 0, "getsockname: ", strerror(errno));

  setNonBlocking(listenFd_);
  loop_->registerDescriptor(listenFd_, EPOLLIN, this);
}

TcpDevice::~TcpDevice() {
  if (listenFd_ >= 0) {
    loop_->unregisterDescriptor(listenFd_);
  }
  // Drop any still-pending accepted connections.
  loop_->defer([this] {
    for (auto& kv : readers_) {
      loop_->unregisterDescriptor(kv.first);
      close(kv.first);
    }
    readers_.clear();
  });
  // Loop destructor (runs after this body) joins the thread; close the
  // listening socket once no more accepts can fire.
  if (listenFd_ >= 0) {
    close(listenFd_);
    listenFd_ = -1;
  }
  for (auto& kv : arrived_) {
    close(kv.second);
  }
}

std::string TcpDevice::str() const {
  return "tcp:" + TcpAddress(advertised_, 0).str();
}

std::shared_ptr<transport::Context> TcpDevice::createContext(
    int rank,
    int size) {
  return std::make_shared<TcpContext>(shared_from_this(), rank, size);
}

void TcpDevice::handleEvents(uint32_t /*events*/) {
  // Accept as many connections as are ready.
  for (;;) {
    int fd = accept4(listenFd_, nullptr, nullptr, SOCK_CLOEXEC);
    if (fd < 0) {
      if (errno == EAGAIN || errno == EWOULDBLOCK) {
        return;
      }
      if (errno == EINTR) {
        continue;
      }
      GA_WARN << "accept: " << strerror(errno);
      return;
    }
    setNonBlocking(fd);
    setSocketOptions(fd);
    auto reader = std::make_unique<SeqReader>(this, fd);
    auto* raw = reader.get();
    {
      std::lock_guard<std::mutex> lock(mu_);
      readers_[fd] = std::move(reader);
    }
    loop_->registerDescriptor(fd, EPOLLIN, raw);
  }
}

void TcpDevice::SeqReader::handleEvents(uint32_t /*events*/) {
  // Loop thread only.
  while (nread_ < sizeof(seq_)) {
    ssize_t n = read(
        fd_, reinterpret_cast<char*>(&seq_) + nread_, sizeof(seq_) - nread_);
    if (n > 0) {
      nread_ += n;
      continue;
    }
    if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      return; // wait for more
    }
    if (n < 0 && errno == EINTR) {
      continue;
    }
    // EOF or error before the preamble: drop the connection.
    dev_->loop().unregisterDescriptor(fd_);
    close(fd_);
    int fd = fd_;
    auto* dev = dev_;
    dev->loop().defer([dev, fd] {
      std::lock_guard<std::mutex> lock(dev->mu_);
      dev->readers_.erase(fd);
    });
    return;
  }
  // Got the full seq: hand off to the routing table. The erase of this
  // SeqReader is deferred to the next loop tick (we are inside its method).
  dev_->loop().unregisterDescriptor(fd_);
  dev_->routeConnection(seq_, fd_);
  int fd = fd_;
  auto* dev = dev_;
  dev->loop().defer([dev, fd] {
    std::lock_guard<std::mutex> lock(dev->mu_);
    dev->readers_.erase(fd);
  });
}

void TcpDevice::routeConnection(uint64_t seq, int fd) {
  std::lock_guard<std::mutex> lock(mu_);
  arrived_[seq] = fd;
  cv_.notify_all();
}

int TcpDevice::waitForConnection(
    uint64_t seq,
    std::chrono::milliseconds timeout) {
  std::unique_lock<std::mutex> lock(mu_);
  auto pred = [&] { return arrived_.count(seq) > 0; };
  if (timeout.count() < 0) {
    cv_.wait(lock, pred);
  } else if (!cv_.wait_for(lock, timeout, pred)) {
    throw TimeoutException(
        "timed out waiting for incoming connection (seq " +
        std::to_string(seq) + ")");
  }
  int fd = arrived_[seq];
  arrived_.erase(seq);
  return fd;
}

} // namespace tcp
} // namespace glooamd
