import os

import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a ROCm GPU (run with -m gpu on an MI355X box)")


os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
# keep synthetic datasets small for CI speed unless a test overrides
os.environ.setdefault("DLB_SYNTH_SCALE", "0.01")


@pytest.fixture
def free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port
