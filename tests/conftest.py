import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)"
    )


@pytest.fixture
def db():
    from kolibrie_amd import SparqlDatabase
    return SparqlDatabase(device="cpu")
