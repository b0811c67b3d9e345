"""ContextFactory: re-rendezvous over a connected context without a
store (reference parity: gloo/rendezvous/context.cc:37-162 +
context_factory_test.cc)."""
import numpy as np

import gloo_amd as ga


def test_context_factory(spawn_threads):
    def fn(ctx, rank, size):
        factory = ga._C.ContextFactory(ctx)
        # Two derived contexts, each fully functional.
        for k in range(2):
            ctx2 = factory.make_context(ga.create_tcp_device())
            x = np.arange(100, dtype=np.float32) + rank + k
            ga.allreduce(ctx2, [x.ctypes.data], 100)
            expected = sum(np.arange(100, dtype=np.float32) + r + k
                           for r in range(size))
            assert np.allclose(x, expected)
        # The backing context still works.
        y = np.full(10, float(rank), dtype=np.float32)
        ga.allreduce(ctx, [y.ctypes.data], 10)
        assert np.allclose(y, sum(range(size)))
        return True

    spawn_threads(3, fn)
