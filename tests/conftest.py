import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


@pytest.fixture
def store():
    from agentcontrolplane_amd.store import ResourceStore

    s = ResourceStore()
    yield s
    s.close()


def wait_for(predicate, timeout=25.0, interval=0.01):
    """Poll until predicate() is truthy; return its value or raise."""
    import time

    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        v = predicate()
        if v:
            return v
        time.sleep(interval)
    raise TimeoutError("condition not met within %.1fs" % timeout)
