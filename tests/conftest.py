import gc

import pytest  # noqa: F401


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture(autouse=True)
def _reap_rpc_cycles():
    """Destroy leaked Rpc machinery promptly between tests.

    Handler closures reference their Rpc wrappers (reference cycles), so a
    test's peers can otherwise survive into the next test until a GC cycle
    runs — keepalive/gossip traffic from those zombies is cross-test
    interference (the reference runs every test --forked for the same
    reason; we reap instead of forking)."""
    yield
    gc.collect()
