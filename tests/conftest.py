import os
import sys

import pytest

# make the repo root importable regardless of pytest invocation dir
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run on the GPU box)")


@pytest.fixture(autouse=True)
def _loopback_master(monkeypatch):
    # container hostnames may not resolve; rendezvous is always loopback
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    yield
