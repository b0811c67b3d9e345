bin/example_allreduce: examples/example_allreduce.cc \
 csrc/algorithms/allreduce_ring.h csrc/algorithms/algorithm.h \
 csrc/context.h csrc/common/store.h csrc/transport/transport.h \
 csrc/math.h csrc/types.h csrc/common/logging.h csrc/common/error.h \
 csrc/common/utils.h csrc/rendezvous/stores.h csrc/transport/tcp/device.h \
 csrc/transport/tcp/address.h csrc/transport/tcp/loop.h
csrc/algorithms/allreduce_ring.h:
csrc/algorithms/algorithm.h:
csrc/context.h:
csrc/common/store.h:
csrc/transport/transport.h:
csrc/math.h:
csrc/types.h:
csrc/common/logging.h:
csrc/common/error.h:
csrc/common/utils.h:
csrc/rendezvous/stores.h:
csrc/transport/tcp/device.h:
csrc/transport/tcp/address.h:
csrc/transport/tcp/loop.h:
