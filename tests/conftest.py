import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (run via gpurun)")


@pytest.fixture(autouse=True)
def _clear_graph():
    from pathway_amd.internals.rungraph import G

    G.clear()
    yield
    G.clear()


def free_port() -> int:
    """An OS-assigned free TCP port (avoids fixed-port collisions when
    several pytest processes run concurrently)."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
