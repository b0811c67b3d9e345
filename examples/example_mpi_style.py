"""MPI-flavored veneer over gloo_amd (reference parity:
gloo/examples/looks_like_mpi.cc): COMM_WORLD-style init from
environment variables, then allreduce/bcast/barrier free functions.

Run: RANK=0 WORLD_SIZE=2 MASTER_PORT=29441 python examples/example_mpi_style.py
(and RANK=1 in a second shell), or just `python ...` for a
single-process world.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import gloo_amd as ga


class Comm:
    def __init__(self):
        self.rank = int(os.environ.get("RANK", "0"))
        self.size = int(os.environ.get("WORLD_SIZE", "1"))
        host = os.environ.get("MASTER_ADDR", "127.0.0.1")
        port = int(os.environ.get("MASTER_PORT", "29441"))
        if self.size > 1:
            store = ga.TcpStore(host, port, is_server=(self.rank == 0))
        else:
            store = ga.HashStore()
        self.ctx = ga.Context(self.rank, self.size)
        self.ctx.connect_full_mesh(store, ga.create_tcp_device())

    def allreduce(self, arr, op=ga.ReduceOp.sum):
        ga.allreduce(self.ctx, [arr.ctypes.data], arr.size,
                     ga.dtype_from_numpy(arr.dtype), op)
        return arr

    def bcast(self, arr, root=0):
        ga.broadcast(self.ctx, arr.ctypes.data, 0, arr.size,
                     ga.dtype_from_numpy(arr.dtype), root=root)
        return arr

    def barrier(self):
        ga.barrier(self.ctx, tag=self.ctx.next_slot())


def main():
    comm = Comm()
    x = np.full(8, float(comm.rank + 1), dtype=np.float32)
    comm.allreduce(x)
    expected = sum(range(1, comm.size + 1))
    assert np.all(x == expected), x
    b = np.arange(4, dtype=np.float32) if comm.rank == 0 else np.zeros(
        4, dtype=np.float32)
    comm.bcast(b, root=0)
    assert np.all(b == np.arange(4, dtype=np.float32))
    comm.barrier()
    print(f"rank {comm.rank}/{comm.size}: allreduce={x[0]:.0f} bcast ok")


if __name__ == "__main__":
    main()
