"""TLS transport tests (reference parity: gloo/test/tls_tcp_test.cc with
self-signed certs)."""
import shutil
import subprocess

import numpy as np
import pytest

import gloo_amd as ga

openssl = shutil.which("openssl")
pytestmark = pytest.mark.skipif(openssl is None, reason="no openssl CLI")


@pytest.fixture(scope="module")
def certs(tmp_path_factory):
    d = tmp_path_factory.mktemp("tls")
    key = d / "key.pem"
    cert = d / "cert.pem"
    subprocess.run(
        [openssl, "req", "-x509", "-newkey", "rsa:2048", "-keyout", str(key),
         "-out", str(cert), "-days", "1", "-nodes", "-subj",
         "/CN=127.0.0.1"],
        check=True, capture_output=True)
    return str(key), str(cert)


def spawn_tls(size, fn, certs, with_ca=False):
    import threading

    key, cert = certs
    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            dev = ga.create_tls_device(
                pkey=key, cert=cert, ca_file=cert if with_ca else "")
            ctx = ga.Context(rank, size)
            ctx.connect_full_mesh(store, dev)
            fn(ctx, rank, size)
            try:
                ga.barrier(ctx, tag=0xFFFF1)
            except ga.GlooAmdError:
                pass
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,)) for r in range(size)]
    [t.start() for t in ths]
    [t.join(60) for t in ths]
    assert not errors, errors[0]


def test_tls_allreduce(certs):
    def fn(ctx, rank, size):
        x = np.arange(5000, dtype=np.float32) + rank
        ga.allreduce(ctx, [x.ctypes.data], x.size)
        expected = sum(np.arange(5000, dtype=np.float32) + r
                       for r in range(size))
        assert np.allclose(x, expected)

    spawn_tls(3, fn, certs)


def test_tls_verified_peer(certs):
    """Self-signed cert doubles as its own CA: peer verification on."""

    def fn(ctx, rank, size):
        x = np.full(100, float(rank), dtype=np.float32)
        ga.allreduce(ctx, [x.ctypes.data], x.size)
        assert np.allclose(x, 1.0)

    spawn_tls(2, fn, certs, with_ca=True)


def test_tls_big_transfer(certs):
    def fn(ctx, rank, size):
        n = 2_000_000
        if rank == 0:
            data = np.arange(n, dtype=np.float32)
            ub = ctx.create_unbound_buffer(data.ctypes.data, data.nbytes)
            ub.send(1, slot=9)
            ub.wait_send()
        else:
            out = np.zeros(n, dtype=np.float32)
            ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
            ub.recv(0, slot=9)
            ub.wait_recv()
            assert np.array_equal(out, np.arange(n, dtype=np.float32))

    spawn_tls(2, fn, certs)


def test_tls_legacy_algorithm(certs):
    def fn(ctx, rank, size):
        x = np.full(1000, float(rank + 1), dtype=np.float32)
        algo = ga._C.create_algorithm("allreduce_ring_chunked", ctx,
                                      [x.ctypes.data], 1000)
        algo.run()
        assert np.allclose(x, 3.0)

    spawn_tls(2, fn, certs)
