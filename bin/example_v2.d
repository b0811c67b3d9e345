bin/example_v2: examples/example_v2.cc csrc/collectives/collectives.h \
 csrc/context.h csrc/common/store.h csrc/transport/transport.h \
 csrc/types.h csrc/common/logging.h csrc/common/error.h \
 csrc/collectives/reduce_fns.h csrc/rendezvous/stores.h \
 csrc/transport/tcp/device.h csrc/transport/tcp/address.h \
 csrc/transport/tcp/loop.h
csrc/collectives/collectives.h:
csrc/context.h:
csrc/common/store.h:
csrc/transport/transport.h:
csrc/types.h:
csrc/common/logging.h:
csrc/common/error.h:
csrc/collectives/reduce_fns.h:
csrc/rendezvous/stores.h:
csrc/transport/tcp/device.h:
csrc/transport/tcp/address.h:
csrc/transport/tcp/loop.h:
