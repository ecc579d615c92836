import socket
import uuid

import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run with -m gpu on a GPU box)"
    )


@pytest.fixture
def ipc_addr(tmp_path):
    """Unique ipc:// address per test (reference tests use ipc://{tmp_path})."""
    return f"ipc://{tmp_path}/{uuid.uuid4().hex[:8]}.ipc"


@pytest.fixture
def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture(autouse=True)
def _quiet_logs(caplog):
    import logging

    logging.getLogger("detectmateservice_amd").setLevel(logging.WARNING)
    yield
