// TLS transport: the tcp transport with OpenSSL framing.
// Capability parity with reference gloo/transport/tcp/tls/ (Device /
// Context / Pair subclasses replacing read/write with SSL_read/SSL_write).
// Re-designed: the handshake runs blocking inside Pair::connect (the
// rendezvous path is already blocking) instead of the reference's
// nonblocking WANT_READ/WANT_WRITE handshake state machine; data-path
// I/O stays nonblocking via the ioRead/ioWritev hooks.
#pragma once

#include <openssl/ssl.h>

#include <mutex>

#include "transport/tcp/context.h"
#include "transport/tcp/device.h"

namespace glooamd {
namespace tcp {
namespace tls {

struct TlsAttr {
  TcpAttr tcp;
  std::string pkeyFile; // PEM private key
  std::string certFile; // PEM certificate
  std::string caFile; // PEM CA bundle (empty -> no peer verification)
  std::string caPath;
};

class TlsDevice;

std::shared_ptr<TlsDevice> createTlsDevice(const TlsAttr& attr);

class TlsDevice : public TcpDevice {
 public:
  explicit TlsDevice(const TlsAttr& attr);
  ~TlsDevice() override;

  std::string str() const override;
  std::shared_ptr<transport::Context> createContext(int rank, int size)
      override;

  SSL_CTX* sslCtx() {
    return sslCtx_;
  }

 private:
  SSL_CTX* sslCtx_{nullptr};
  bool verifyPeer_{false};
};

class TlsPair : public TcpPair {
 public:
  TlsPair(TcpContext* ctx, TlsDevice* dev, int peerRank)
      : TcpPair(ctx, dev, peerRank),
        sslCtx_(dev->sslCtx()),
        handshakeTimeout_(ctx->getTimeout()) {}
  ~TlsPair() override;

 protected:
  ssize_t ioRead(char* buf, size_t len) override;
  ssize_t ioWritev(const struct iovec* iov, int iovcnt) override;
  void ioHandshake(bool initiator) override;
  void ioClose() override;

 private:
  SSL_CTX* sslCtx_;
  SSL* ssl_{nullptr};
  std::chrono::milliseconds handshakeTimeout_;
  // OpenSSL SSL objects are not thread-safe: the loop thread reads while
  // a user thread may be flushing writes, so every SSL_* data call is
  // serialized per pair. Lock order is ctx-mutex -> sslMu_ (writes) or
  // sslMu_ alone (payload reads); the read path never takes the ctx
  // mutex while holding sslMu_, so the order is consistent.
  std::mutex sslMu_;
};

class TlsContext : public TcpContext {
 public:
  TlsContext(std::shared_ptr<TlsDevice> device, int rank, int size)
      : TcpContext(device, rank, size) {}

  transport::Pair* createPair(int rank) override;
};

} // namespace tls
} // namespace tcp
} // namespace glooamd
