import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (run via gpurun)")


@pytest.fixture(autouse=True)
def _clear_graph():
    from pathway_amd.internals.rungraph import G

    G.clear()
    yield
    G.clear()
