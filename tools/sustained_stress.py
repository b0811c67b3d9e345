"""Sustained-concurrency liveness stress (threaded ranks, fixed
iteration counts — collectives must be invoked equally on every rank).

Usage: python tools/sustained_stress.py <ranks> <comma-sizes>
e.g.   python tools/sustained_stress.py 8 64,3000,50000,400000
Env:   GLOO_AMD_UDS=1 to run over unix sockets.
"""
import os
import sys
import threading
import time  # noqa: F401
sys.path.insert(0, __file__.rsplit("/", 2)[0])
import numpy as np
import gloo_amd as ga

size = int(sys.argv[1]); sizes = [int(x) for x in sys.argv[2].split(",")]
dur = float(sys.argv[3]) if len(sys.argv) > 3 else 8
store = ga.HashStore(); errors = []

def worker(rank):
    try:
        dev = ga.create_tcp_device(
            use_uds=os.environ.get("GLOO_AMD_UDS") == "1")
        ctx = ga.Context(rank, size)
        ctx.connect_full_mesh(store, dev)
        ctx.set_timeout(20000)
        iters = {64: 400, 3000: 300, 50_000: 150, 400_000: 60}
        def stream(tag, n):
            for it in range(iters.get(n, 100)):
                x = (np.arange(n, dtype=np.float64) + rank + it).astype(np.float32)
                ga.allreduce(ctx, [x.ctypes.data], n, ga.DType.f32,
                             ga.ReduceOp.sum, tag=tag)
                ref = sum((np.arange(n, dtype=np.float64) + r + it).astype(np.float32)
                          for r in range(size))
                assert np.allclose(x, ref), (tag, it)
        ths = [threading.Thread(target=stream, args=(t + 1, n))
               for t, n in enumerate(sizes)]
        [t.start() for t in ths]
        [t.join() for t in ths]
        ga.barrier(ctx, tag=999)
    except Exception:
        import traceback
        errors.append(traceback.format_exc())

ths = [threading.Thread(target=worker, args=(r,)) for r in range(size)]
[t.start() for t in ths]
[t.join(90) for t in ths]
print("FAIL" if errors else "OK", len(errors), "errors")
sys.exit(1 if errors else 0)
