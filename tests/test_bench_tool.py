"""Benchmark binary smoke tests (reference parity: gloo/benchmark)."""
import os
import subprocess

import pytest

BIN = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "bin", "gloo_amd_bench")

pytestmark = pytest.mark.skipif(
    not os.path.exists(BIN), reason="bench binary not built")


def run_bench(size, benchmark, elements, extra=None, timeout=90):
    import tempfile

    with tempfile.TemporaryDirectory() as store:
        procs = [
            subprocess.Popen(
                [BIN, "--size", str(size), "--rank", str(r),
                 "--store-file", store, "--benchmark", benchmark,
                 "--elements", str(elements), "--iteration-time-ms", "100"]
                + (extra or []),
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
            for r in range(size)
        ]
        outs = [p.communicate(timeout=timeout)[0].decode() for p in procs]
        assert all(p.returncode == 0 for p in procs), "\n".join(outs)
        return outs


@pytest.mark.parametrize("benchmark", [
    "allreduce_ring", "allreduce_ring_chunked", "allreduce_halving_doubling",
    "allreduce_bcube", "new_allreduce_ring", "allgather_ring",
    "barrier_all_to_all", "broadcast_one_to_all", "pairwise_exchange",
    "reduce_scatter_halving_doubling", "sendrecv_roundtrip",
])
def test_bench_cpu(benchmark):
    outs = run_bench(2, benchmark, 1000)
    assert any("p50" in o for o in outs)


def test_bench_halving_doubling_3ranks():
    run_bench(3, "allreduce_halving_doubling", 5000)


def test_bench_half_precision():
    run_bench(2, "allreduce_ring_chunked", 1000, extra=["--half-precision"])
