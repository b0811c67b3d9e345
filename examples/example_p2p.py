"""Pipeline-parallel-shaped point-to-point over gloo_amd: tagged
send/recv between stages plus recv-from-any for the collector rank.

Run: python examples/example_p2p.py  (spawns 3 worker threads)
Counterpart of the reference's send/recv usage (gloo/examples and
gloo/test/send_recv_test.cc patterns), re-done over this API.
"""
import sys
import os
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import gloo_amd as ga

WORLD = 3
STEPS = 4


def stage(ctx, rank):
    if rank < WORLD - 1:
        # pipeline stage: transform and pass activations downstream
        for step in range(STEPS):
            act = np.full(1024, float(rank * 100 + step), dtype=np.float32)
            if rank > 0:
                inp = np.zeros(1024, dtype=np.float32)
                ub = ctx.create_unbound_buffer(inp.ctypes.data, inp.nbytes)
                ub.recv(rank - 1, slot=step)
                ub.wait_recv()
                act += inp
            ub = ctx.create_unbound_buffer(act.ctypes.data, act.nbytes)
            ub.send(rank + 1, slot=step)
            ub.wait_send()
    else:
        # collector: receive each step from whichever stage finishes
        got = 0
        while got < STEPS:
            out = np.zeros(1024, dtype=np.float32)
            ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
            ub.recv(rank - 1, slot=got)
            ok, src = ub.wait_recv()
            assert ok
            print(f"collector got step {got} from rank {src}: {out[0]:.0f}")
            got += 1


def main():
    store = ga.HashStore()

    def worker(rank):
        dev = ga.create_tcp_device()
        ctx = ga.Context(rank, WORLD)
        ctx.connect_full_mesh(store, dev)
        stage(ctx, rank)

    threads = [threading.Thread(target=worker, args=(r,)) for r in range(WORLD)]
    [t.start() for t in threads]
    [t.join() for t in threads]
    print("p2p example done")


if __name__ == "__main__":
    main()
