#include "transport/tcp/tls.h"

#include <fcntl.h>
#include <openssl/err.h>
#include <poll.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>

#include "common/logging.h"

namespace glooamd {
namespace tcp {
namespace tls {

namespace {
std::string sslErrors() {
  std::string out;
  unsigned long e;
  while ((e = ERR_get_error()) != 0) {
    char buf[256];
    ERR_error_string_n(e, buf, sizeof(buf));
    out += std::string(buf) + "; ";
  }
  return out;
}

void setBlocking(int fd, bool blocking) {
  int flags = fcntl(fd, F_GETFL, 0);
  if (blocking) {
    fcntl(fd, F_SETFL, flags & ~O_NONBLOCK);
  } else {
    fcntl(fd, F_SETFL, flags | O_NONBLOCK);
  }
}
} // namespace

std::shared_ptr<TlsDevice> createTlsDevice(const TlsAttr& attr) {
  return std::make_shared<TlsDevice>(attr);
}

TlsDevice::TlsDevice(const TlsAttr& attr) : TcpDevice(attr.tcp) {
  static std::once_flag once;
  std::call_once(once, [] {
    SSL_library_init();
    SSL_load_error_strings();
  });
  sslCtx_ = SSL_CTX_new(TLS_method());
  GA_ENFORCE(sslCtx_ != nullptr, "SSL_CTX_new: ", sslErrors());
  SSL_CTX_set_min_proto_version(sslCtx_, TLS1_2_VERSION);
  // Partial writes + moving buffers: our tx queue retries writev-style.
  SSL_CTX_set_mode(
      sslCtx_,
      SSL_MODE_ENABLE_PARTIAL_WRITE | SSL_MODE_ACCEPT_MOVING_WRITE_BUFFER);
  GA_ENFORCE(
      SSL_CTX_use_certificate_file(
          sslCtx_, attr.certFile.c_str(), SSL_FILETYPE_PEM) == 1,
      "cert load failed: ",
      sslErrors());
  GA_ENFORCE(
      SSL_CTX_use_PrivateKey_file(
          sslCtx_, attr.pkeyFile.c_str(), SSL_FILETYPE_PEM) == 1,
      "key load failed: ",
      sslErrors());
  if (!attr.caFile.empty() || !attr.caPath.empty()) {
    GA_ENFORCE(
        SSL_CTX_load_verify_locations(
            sslCtx_,
            attr.caFile.empty() ? nullptr : attr.caFile.c_str(),
            attr.caPath.empty() ? nullptr : attr.caPath.c_str()) == 1,
        "CA load failed: ",
        sslErrors());
    SSL_CTX_set_verify(
        sslCtx_, SSL_VERIFY_PEER | SSL_VERIFY_FAIL_IF_NO_PEER_CERT, nullptr);
    verifyPeer_ = true;
  } else {
    SSL_CTX_set_verify(sslCtx_, SSL_VERIFY_NONE, nullptr);
  }
}

TlsDevice::~TlsDevice() {
  if (sslCtx_ != nullptr) {
    SSL_CTX_free(sslCtx_);
  }
}

std::string TlsDevice::str() const {
  return "tls+" + TcpDevice::str();
}

std::shared_ptr<transport::Context> TlsDevice::createContext(
    int rank,
    int size) {
  return std::make_shared<TlsContext>(
      std::static_pointer_cast<TlsDevice>(shared_from_this()), rank, size);
}

transport::Pair* TlsContext::createPair(int rank) {
  GA_ENFORCE_NE(rank, this->rank, "no pair to self");
  setPair(
      rank,
      std::make_unique<TlsPair>(
          this, static_cast<TlsDevice*>(device()), rank));
  return getPair(rank);
}

TlsPair::~TlsPair() {
  // Base destructor calls close() -> ioClose() while ssl_ is alive.
  close();
  if (ssl_ != nullptr) {
    SSL_free(ssl_);
    ssl_ = nullptr;
  }
}

void TlsPair::ioHandshake(bool initiator) {
  ssl_ = SSL_new(sslCtx_);
  GA_ENFORCE(ssl_ != nullptr, "SSL_new: ", sslErrors());
  // Nonblocking handshake with a deadline (reference role:
  // gloo/transport/tcp/tls/pair.cc:44-170 SSL_do_handshake state
  // machine): SSL_do_handshake on a nonblocking fd, polling for the
  // direction OpenSSL asks for, bounded by the pair timeout — a peer
  // that stalls mid-handshake can no longer hang connect() forever.
  setBlocking(fd(), false);
  GA_ENFORCE_EQ(SSL_set_fd(ssl_, fd()), 1, "SSL_set_fd: ", sslErrors());
  if (initiator) {
    SSL_set_connect_state(ssl_);
  } else {
    SSL_set_accept_state(ssl_);
  }
  auto timeout = handshakeTimeout_;
  if (timeout.count() <= 0) {
    timeout = std::chrono::milliseconds(30000);
  }
  const auto deadline = std::chrono::steady_clock::now() + timeout;
  for (;;) {
    int rv = SSL_do_handshake(ssl_);
    if (rv == 1) {
      return;
    }
    int err = SSL_get_error(ssl_, rv);
    if (err != SSL_ERROR_WANT_READ && err != SSL_ERROR_WANT_WRITE) {
      GA_THROW_IO(
          "TLS handshake failed (", initiator ? "connect" : "accept",
          ", err=", err, "): ", sslErrors());
    }
    const auto now = std::chrono::steady_clock::now();
    if (now >= deadline) {
      GA_THROW_IO(
          "TLS handshake timed out (", initiator ? "connect" : "accept",
          ")");
    }
    struct pollfd pfd;
    pfd.fd = fd();
    pfd.events = (err == SSL_ERROR_WANT_READ) ? POLLIN : POLLOUT;
    pfd.revents = 0;
    const int waitMs = static_cast<int>(std::min<int64_t>(
        std::chrono::duration_cast<std::chrono::milliseconds>(
            deadline - now)
            .count(),
        100));
    (void)::poll(&pfd, 1, std::max(waitMs, 1));
  }
}

void TlsPair::ioClose() {
  if (ssl_ != nullptr) {
    (void)SSL_shutdown(ssl_); // best-effort close_notify
  }
}

ssize_t TlsPair::ioRead(char* buf, size_t len) {
  std::lock_guard<std::mutex> lock(sslMu_);
  int n = SSL_read(ssl_, buf, static_cast<int>(std::min<size_t>(len, 1 << 30)));
  if (n > 0) {
    return n;
  }
  int err = SSL_get_error(ssl_, n);
  switch (err) {
    case SSL_ERROR_WANT_READ:
    case SSL_ERROR_WANT_WRITE:
      errno = EAGAIN;
      return -1;
    case SSL_ERROR_ZERO_RETURN:
      return 0; // clean TLS EOF
    case SSL_ERROR_SYSCALL:
      if (errno == 0) {
        return 0; // peer closed without close_notify
      }
      return -1;
    default:
      errno = EIO;
      return -1;
  }
}

ssize_t TlsPair::ioWritev(const struct iovec* iov, int iovcnt) {
  std::lock_guard<std::mutex> lock(sslMu_);
  ssize_t total = 0;
  for (int i = 0; i < iovcnt; i++) {
    size_t off = 0;
    while (off < iov[i].iov_len) {
      int n = SSL_write(
          ssl_,
          static_cast<const char*>(iov[i].iov_base) + off,
          static_cast<int>(
              std::min<size_t>(iov[i].iov_len - off, 1 << 30)));
      if (n > 0) {
        off += n;
        total += n;
        continue;
      }
      int err = SSL_get_error(ssl_, n);
      if (err == SSL_ERROR_WANT_WRITE || err == SSL_ERROR_WANT_READ) {
        if (total > 0) {
          return total; // partial progress; caller re-arms EPOLLOUT
        }
        errno = EAGAIN;
        return -1;
      }
      if (total > 0) {
        return total;
      }
      errno = EIO;
      return -1;
    }
  }
  return total;
}

} // namespace tls
} // namespace tcp
} // namespace glooamd
