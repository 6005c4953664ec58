import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    # Nothing needed: -m "not gpu" / -m gpu filtering is done by the driver.
    pass


@pytest.fixture
def kv_server():
    from tf_yarn_amd.kv import KVServer
    server = KVServer()
    yield server
    server.stop()


@pytest.fixture
def kv_client(kv_server):
    from tf_yarn_amd.kv import KVClient
    client = KVClient(kv_server.address)
    yield client
    client.close()
