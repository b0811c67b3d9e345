// v2 function-collectives example (reference parity:
// gloo/examples/example_allreduce_v2.cc): tcp store rendezvous,
// allreduce + tagged send/recv.
//
//   ./example_v2 <rank> <size> <host:port>   (rank 0 hosts the store)
#include <cstdio>
#include <cstring>
#include <vector>

#include "collectives/collectives.h"
#include "collectives/reduce_fns.h"
#include "context.h"
#include "rendezvous/stores.h"
#include "transport/tcp/device.h"

int main(int argc, char** argv) {
  if (argc != 4) {
    fprintf(stderr, "usage: %s <rank> <size> <host:port>\n", argv[0]);
    return 1;
  }
  const int rank = atoi(argv[1]);
  const int size = atoi(argv[2]);
  std::string hp = argv[3];
  auto colon = hp.find(':');

  glooamd::TcpStore store(
      hp.substr(0, colon), atoi(hp.c_str() + colon + 1), rank == 0);
  auto device = glooamd::tcp::createTcpDevice();
  auto context = std::make_shared<glooamd::Context>(rank, size);
  context->connectFullMesh(store, device);

  // Allreduce.
  std::vector<float> data(8, float(rank + 1));
  glooamd::AllreduceOptions opts(context);
  opts.setOutput(data.data(), data.size());
  opts.reduce = glooamd::cpuReduceFn(
      glooamd::DType::F32, glooamd::ReduceOp::SUM);
  glooamd::allreduce(opts);
  printf("rank %d allreduce -> %g\n", rank, data[0]);

  // Tagged point-to-point ring: pass a token around.
  char token[32];
  auto buf = context->createUnboundBuffer(token, sizeof(token));
  const uint64_t slot =
      glooamd::Slot::build(glooamd::SlotPrefix::kSendRecv, 1);
  if (rank == 0) {
    snprintf(token, sizeof(token), "hello from 0");
    buf->send((rank + 1) % size, slot);
    buf->waitSend();
    buf->recv((rank - 1 + size) % size, slot);
    buf->waitRecv();
  } else {
    buf->recv((rank - 1 + size) % size, slot);
    buf->waitRecv();
    buf->send((rank + 1) % size, slot);
    buf->waitSend();
  }
  printf("rank %d token: %s\n", rank, token);
  return 0;
}
