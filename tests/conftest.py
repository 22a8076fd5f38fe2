import os
import tempfile

import pytest
import torch

# keep test-generated profiler artifacts out of the committed profiles/
os.environ.setdefault("DDLB_PROFILE_DIR",
                      os.path.join(tempfile.gettempdir(),
                                   "ddlb_test_profiles"))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a HIP GPU (run on the MI355X box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no HIP device in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port
