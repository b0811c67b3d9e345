// Minimal legacy-API example (reference parity: gloo/examples/example1.cc):
// FileStore rendezvous + AllreduceRing over the tcp transport.
//
//   ./example_allreduce <rank> <size> <store-dir>
#include <cstdio>
#include <vector>

#include "algorithms/allreduce_ring.h"
#include "context.h"
#include "rendezvous/stores.h"
#include "transport/tcp/device.h"

int main(int argc, char** argv) {
  if (argc != 4) {
    fprintf(stderr, "usage: %s <rank> <size> <store-dir>\n", argv[0]);
    return 1;
  }
  const int rank = atoi(argv[1]);
  const int size = atoi(argv[2]);

  glooamd::FileStore store(argv[3]);
  auto device = glooamd::tcp::createTcpDevice();
  auto context = std::make_shared<glooamd::Context>(rank, size);
  context->connectFullMesh(store, device);

  std::vector<float> data(16);
  for (size_t i = 0; i < data.size(); i++) {
    data[i] = rank * 100 + i;
  }
  glooamd::AllreduceRing<float> allreduce(
      context, {data.data()}, data.size());
  allreduce.run();

  printf("rank %d:", rank);
  for (float v : data) {
    printf(" %g", v);
  }
  printf("\n");
  return 0;
}
