// TCP address = advertised sockaddr + 64-bit pair sequence number.
// Capability parity with reference gloo/transport/tcp/address.h:27-84.
// The seq number routes an incoming connection on the device's single
// listening socket to the pair that owns it.
#pragma once

#include <netinet/in.h>
#include <sys/socket.h>

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

#include "transport/transport.h"

namespace glooamd {
namespace tcp {

class TcpAddress : public transport::Address {
 public:
  TcpAddress() {
    std::memset(&ss_, 0, sizeof(ss_));
    seq_ = 0;
  }
  TcpAddress(const struct sockaddr_storage& ss, uint64_t seq)
      : ss_(ss), seq_(seq) {}

  // Deserialize from bytes() output.
  static TcpAddress fromBytes(const std::vector<char>& bytes);

  std::vector<char> bytes() const override;
  std::string str() const override;

  const struct sockaddr_storage& sockaddr() const {
    return ss_;
  }
  socklen_t sockaddrLen() const;
  uint64_t seq() const {
    return seq_;
  }
  void setSeq(uint64_t seq) {
    seq_ = seq;
  }

 private:
  struct sockaddr_storage ss_;
  uint64_t seq_;
};

} // namespace tcp
} // namespace glooamd
