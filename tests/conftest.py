import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


def pytest_collection_modifyitems(items):
    """Safety net: a wedged GPU protocol must FAIL the run, not hang it
    forever (pytest-timeout 'thread' method dumps all stacks and exits).
    The per-test watchdogs normally fire long before this."""
    try:
        import pytest_timeout  # noqa: F401
    except ImportError:
        return
    for item in items:
        if item.get_closest_marker("gpu") is not None \
                and item.get_closest_marker("timeout") is None:
            item.add_marker(pytest.mark.timeout(420, method="thread"))


@pytest.fixture
def spawn_threads():
    """gloo-style multi-rank harness: N python threads, each with its own
    tcp Device + Context, meeting through a shared in-process HashStore
    (reference strategy: gloo/test/base_test.h:89-192)."""
    import threading

    import gloo_amd as ga

    def run(size, fn, base=2, timeout=60.0):
        store = ga.HashStore()
        results = [None] * size
        errors = []

        def worker(rank):
            try:
                dev = ga.create_tcp_device()
                ctx = ga.Context(rank, size, base)
                ctx.connect_full_mesh(store, dev)
                results[rank] = fn(ctx, rank, size)
                # Sync before teardown so no rank closes its pairs while a
                # peer is still mid-collective (reference strategy:
                # gloo/test/base_test.h:161-173). Best-effort: tests that
                # deliberately poison the context (timeouts) skip it.
                try:
                    ga.barrier(ctx, tag=0xFFFF0)
                except ga.GlooAmdError:
                    pass
            except Exception as e:  # noqa: BLE001
                import traceback

                errors.append((rank, e, traceback.format_exc()))

        threads = [
            threading.Thread(target=worker, args=(r,), daemon=True)
            for r in range(size)
        ]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout)
            assert not t.is_alive(), f"worker thread hung (>{timeout}s)"
        assert not errors, f"rank errors: {errors[0][2]}"
        return results

    return run
