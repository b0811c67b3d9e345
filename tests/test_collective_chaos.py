"""Randomized collective-sequence chaos: every rank executes the same
seeded script of mixed v2 collectives (random ops, sizes straddling the
eager/small-allreduce/segment thresholds, random algorithms and tags)
and verifies every result. Complements test_protocol_fuzz (p2p) by
exercising cross-collective state: slot allocation, eager stash
handoffs between ops, and ring/bcube/recursive-doubling interleaving on
one context."""
import random

import numpy as np
import pytest

import gloo_amd as ga

SIZES = [1, 3, 100, 4095, 4096, 4097, 10_000, 100_000, 262_145]


def _script(seed, world, n_ops):
    rng = random.Random(seed)
    ops = []
    for i in range(n_ops):
        kind = rng.choice(
            ["allreduce", "allreduce", "allreduce_bcube", "allgather",
             "reduce_scatter", "alltoall", "broadcast", "barrier"])
        ops.append({
            "i": i,
            "kind": kind,
            "n": rng.choice(SIZES),
            "root": rng.randrange(world),
            "tag": 1000 + i,
        })
    return ops


@pytest.mark.parametrize("world,seed", [(2, 7), (3, 13), (4, 29)])
def test_collective_chaos(spawn_threads, world, seed):
    ops = _script(seed, world, 40)

    def fill(rank, n, i):
        return ((np.arange(n, dtype=np.float32) % 13)
                + rank * 3 + i % 5).astype(np.float32)

    def fn(ctx, rank, _):
        for op in ops:
            n, i, root, tag = op["n"], op["i"], op["root"], op["tag"]
            if op["kind"] in ("allreduce", "allreduce_bcube"):
                x = fill(rank, n, i)
                algo = ("bcube" if op["kind"] == "allreduce_bcube"
                        else "ring")
                ga.allreduce(ctx, [x.ctypes.data], n, ga.DType.f32,
                             ga.ReduceOp.sum, tag=tag, algorithm=algo)
                expect = sum(fill(r, n, i) for r in range(world))
                assert np.allclose(x, expect), (op, rank)
            elif op["kind"] == "allgather":
                inp = fill(rank, n, i)
                out = np.zeros(n * world, dtype=np.float32)
                ga.allgather(ctx, out.ctypes.data, inp.ctypes.data, n,
                             ga.DType.f32, tag=tag)
                for r in range(world):
                    assert np.array_equal(out[r * n:(r + 1) * n],
                                          fill(r, n, i)), (op, rank, r)
            elif op["kind"] == "reduce_scatter":
                inp = fill(rank, n * world, i)
                out = np.zeros(n, dtype=np.float32)
                ga.reduce_scatter(ctx, out.ctypes.data, inp.ctypes.data,
                                  n, ga.DType.f32, ga.ReduceOp.sum,
                                  tag=tag)
                total = sum(fill(r, n * world, i) for r in range(world))
                assert np.allclose(
                    out, total[rank * n:(rank + 1) * n]), (op, rank)
            elif op["kind"] == "alltoall":
                inp = fill(rank, n * world, i)
                out = np.zeros(n * world, dtype=np.float32)
                ga.alltoall(ctx, out.ctypes.data, inp.ctypes.data, n,
                            ga.DType.f32, tag=tag)
                for s in range(world):
                    assert np.array_equal(
                        out[s * n:(s + 1) * n],
                        fill(s, n * world, i)[rank * n:(rank + 1) * n]), (
                        op, rank, s)
            elif op["kind"] == "broadcast":
                x = (fill(root, n, i) if rank == root
                     else np.zeros(n, dtype=np.float32))
                ga.broadcast(ctx, x.ctypes.data, 0, n, ga.DType.f32,
                             root=root, tag=tag)
                assert np.array_equal(x, fill(root, n, i)), (op, rank)
            else:
                ga.barrier(ctx, tag=tag)
        return True

    assert all(spawn_threads(world, fn))
