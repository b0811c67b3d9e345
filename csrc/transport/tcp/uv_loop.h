// libuv-driven event loop for the tcp transport.
// Capability parity with the reference's uv transport
// (gloo/transport/uv/): same wire protocol and pair state machine,
// driven by a libuv loop (uv_poll watchers over the pair sockets)
// instead of raw epoll. Select with TcpAttr::useLibuv.
#pragma once

#include <memory>

#include "transport/tcp/loop.h"

namespace glooamd {
namespace tcp {

// Factory keeps <uv.h> out of the public headers.
std::unique_ptr<Loop> makeUvLoop();

} // namespace tcp
} // namespace glooamd
