#include "transport/tcp/address.h"

#include <arpa/inet.h>
#include <sys/un.h>

#include "common/logging.h"

namespace glooamd {
namespace tcp {

std::vector<char> TcpAddress::bytes() const {
  std::vector<char> out(sizeof(ss_) + sizeof(seq_));
  std::memcpy(out.data(), &ss_, sizeof(ss_));
  std::memcpy(out.data() + sizeof(ss_), &seq_, sizeof(seq_));
  return out;
}

TcpAddress TcpAddress::fromBytes(const std::vector<char>& bytes) {
  GA_ENFORCE_EQ(
      bytes.size(),
      sizeof(struct sockaddr_storage) + sizeof(uint64_t),
      "malformed tcp address");
  TcpAddress addr;
  std::memcpy(&addr.ss_, bytes.data(), sizeof(addr.ss_));
  std::memcpy(&addr.seq_, bytes.data() + sizeof(addr.ss_), sizeof(addr.seq_));
  return addr;
}

socklen_t TcpAddress::sockaddrLen() const {
  switch (ss_.ss_family) {
    case AF_INET:
      return sizeof(struct sockaddr_in);
    case AF_INET6:
      return sizeof(struct sockaddr_in6);
    case AF_UNIX:
      return sizeof(struct sockaddr_un); // abstract name, zero-padded
    default:
      return sizeof(ss_);
  }
}

std::string TcpAddress::str() const {
  char host[INET6_ADDRSTRLEN] = {0};
  int port = 0;
  if (ss_.ss_family == AF_UNIX) {
    return std::string("uds#") + std::to_string(seq_);
  }
  if (ss_.ss_family == AF_INET) {
    auto* in = reinterpret_cast<const struct sockaddr_in*>(&ss_);
    inet_ntop(AF_INET, &in->sin_addr, host, sizeof(host));
    port = ntohs(in->sin_port);
  } else if (ss_.ss_family == AF_INET6) {
    auto* in6 = reinterpret_cast<const struct sockaddr_in6*>(&ss_);
    inet_ntop(AF_INET6, &in6->sin6_addr, host, sizeof(host));
    port = ntohs(in6->sin6_port);
  }
  return std::string(host) + ":" + std::to_string(port) + "#" +
      std::to_string(seq_);
}

} // namespace tcp
} // namespace glooamd
