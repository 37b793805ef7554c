import pytest  # noqa: F401


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X (or any ROCm) GPU"
    )


@pytest.fixture(params=[False, True], ids=["batching_on", "batching_off"])
def toggle_batching(request, monkeypatch):
    """Run the decorated e2e test with slab batching enabled and disabled."""
    monkeypatch.setenv("TSAMD_DISABLE_BATCHING", "1" if request.param else "0")
    yield request.param


@pytest.fixture(params=[False, True], ids=["chunking_off", "chunking_on"])
def toggle_chunking(request, monkeypatch):
    """Run the decorated e2e test with a tiny chunk size to force chunking."""
    if request.param:
        monkeypatch.setenv("TSAMD_MAX_CHUNK_SIZE_BYTES", "1024")
    yield request.param
