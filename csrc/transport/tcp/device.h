// TCP Device: owns the epoll Loop thread + the listening socket and routes
// accepted connections to pairs by sequence number.
// Capability parity with reference gloo/transport/tcp/device.cc:259-397 and
// listener.cc:57-146. Re-designed: listener logic folded into the Device,
// rank-ordered connect initiation (lower rank dials) replaces address
// comparison, and connections carry an 8-byte little-endian seq preamble.
#pragma once

#include <atomic>
#include <condition_variable>
#include <memory>
#include <mutex>
#include <unordered_map>

#include "transport/tcp/address.h"
#include "transport/tcp/loop.h"
#include "transport/transport.h"

namespace glooamd {
namespace tcp {

struct TcpAttr {
  // Address to advertise to peers. Empty -> 127.0.0.1 (single-node
  // default; container hostnames may not resolve).
  std::string hostname;
  // Interface to bind (optional, unused when hostname given).
  std::string iface;
  int port = 0; // 0 -> ephemeral
  // Drive the transport with a libuv loop instead of raw epoll
  // (reference uv-transport parity; same wire protocol).
  bool useLibuv = false;
  // Unix-domain stream sockets (abstract namespace) instead of TCP:
  // same wire protocol, lower latency and higher throughput for the
  // single-node case. Peers must share a kernel.
  bool useUds = false;
};

class TcpDevice;

std::shared_ptr<TcpDevice> createTcpDevice(const TcpAttr& attr = TcpAttr());

class TcpDevice : public transport::Device,
                  public Handler,
                  public std::enable_shared_from_this<TcpDevice> {
 public:
  explicit TcpDevice(const TcpAttr& attr);
  ~TcpDevice() override;

  std::string str() const override;
  std::shared_ptr<transport::Context> createContext(int rank, int size)
      override;

  Loop& loop() {
    return *loop_;
  }

  uint64_t nextSeq() {
    return seq_.fetch_add(1);
  }

  // Advertised address (host:port) with a given pair seq.
  TcpAddress addressForSeq(uint64_t seq) const {
    return TcpAddress(advertised_, seq);
  }

  // Listener side of connect: block until a connection carrying `seq`
  // arrives (or throw TimeoutException).
  int waitForConnection(uint64_t seq, std::chrono::milliseconds timeout);

  // Loop-thread callbacks.
  void handleEvents(uint32_t events) override; // accept
  void routeConnection(uint64_t seq, int fd);

 private:
  // Reads the 8-byte seq preamble off a freshly accepted connection on the
  // loop thread, then routes it.
  class SeqReader : public Handler {
   public:
    SeqReader(TcpDevice* dev, int fd) : dev_(dev), fd_(fd) {}
    void handleEvents(uint32_t events) override;

   private:
    TcpDevice* dev_;
    int fd_;
    uint64_t seq_{0};
    size_t nread_{0};
  };

  int listenFd_{-1};
  struct sockaddr_storage advertised_; // host:port peers should dial
  std::atomic<uint64_t> seq_{0};

  std::mutex mu_;
  std::condition_variable cv_;
  std::unordered_map<uint64_t, int> arrived_; // seq -> connected fd
  std::unordered_map<int, std::unique_ptr<SeqReader>> readers_;

  // declared last: destroyed first, joining the loop thread
  std::unique_ptr<Loop> loop_;
};

// Socket helpers shared by pair/device.
void setNonBlocking(int fd);
void setSocketOptions(int fd);

} // namespace tcp
} // namespace glooamd
