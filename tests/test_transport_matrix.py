"""Core collectives over every transport (reference parity: gtest
Combine over transport x ranks x elements, gloo/test/base_test.h:65-77
TCP / TCP_TLS / UV parameterization)."""
import shutil
import subprocess
import threading

import numpy as np
import pytest

import gloo_amd as ga

openssl = shutil.which("openssl")


@pytest.fixture(scope="module")
def tls_certs(tmp_path_factory):
    if openssl is None:
        return None
    d = tmp_path_factory.mktemp("tlsmx")
    key, cert = d / "key.pem", d / "cert.pem"
    subprocess.run(
        [openssl, "req", "-x509", "-newkey", "rsa:2048", "-keyout", str(key),
         "-out", str(cert), "-days", "1", "-nodes", "-subj", "/CN=127.0.0.1"],
        check=True, capture_output=True)
    return str(key), str(cert)


def _spawn(size, fn, transport, tls_certs):
    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            if transport == "tcp":
                dev = ga.create_tcp_device()
            elif transport == "uds":
                dev = ga.create_tcp_device(use_uds=True)
            elif transport == "uv":
                dev = ga.create_tcp_device(use_libuv=True)
            else:
                key, cert = tls_certs
                dev = ga.create_tls_device(pkey=key, cert=cert)
            ctx = ga.Context(rank, size)
            ctx.connect_full_mesh(store, dev)
            fn(ctx, rank, size)
            try:
                ga.barrier(ctx, tag=0xFFFF0)
            except ga.GlooAmdError:
                pass
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,), daemon=True)
           for r in range(size)]
    [t.start() for t in ths]
    [t.join(60) for t in ths]
    assert not any(t.is_alive() for t in ths), "hung"
    assert not errors, errors[0]


TRANSPORTS = ["tcp", "uds", "uv"] + (["tls"] if openssl else [])


@pytest.mark.parametrize("transport", TRANSPORTS)
@pytest.mark.parametrize("size", [2, 4])
def test_allreduce_matrix(transport, size, tls_certs):
    def fn(ctx, rank, _):
        x = (np.arange(3000, dtype=np.float64) + rank).astype(np.float32)
        ga.allreduce(ctx, [x.ctypes.data], x.size, ga.DType.f32,
                     ga.ReduceOp.sum)
        ref = sum((np.arange(3000, dtype=np.float64) + r).astype(np.float32)
                  for r in range(size))
        assert np.allclose(x, ref)

    _spawn(size, fn, transport, tls_certs)


@pytest.mark.parametrize("transport", TRANSPORTS)
def test_sendrecv_any_matrix(transport, tls_certs):
    size = 3

    def fn(ctx, rank, _):
        if rank in (0, 1):
            v = np.full(64, rank + 1.0, dtype=np.float32)
            ub = ctx.create_unbound_buffer(v.ctypes.data, v.nbytes)
            ub.send(2, slot=5)
            ub.wait_send()
        else:
            seen = set()
            for _ in range(2):
                out = np.zeros(64, dtype=np.float32)
                ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
                ub.recv_any([0, 1], slot=5)
                ok, src = ub.wait_recv()
                assert ok and np.all(out == src + 1.0)
                seen.add(src)
            assert seen == {0, 1}

    _spawn(size, fn, transport, tls_certs)


@pytest.mark.parametrize("transport", TRANSPORTS)
def test_legacy_chunked_matrix(transport, tls_certs):
    size = 2

    def fn(ctx, rank, _):
        x = (np.arange(5000, dtype=np.float64) + rank).astype(np.float32)
        a = ga._C.create_algorithm("allreduce_ring_chunked", ctx,
                                   [x.ctypes.data], 5000)
        a.run()
        ref = sum((np.arange(5000, dtype=np.float64) + r).astype(np.float32)
                  for r in range(size))
        assert np.allclose(x, ref)

    _spawn(size, fn, transport, tls_certs)
