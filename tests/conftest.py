import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need an MI355X GPU (run via gpurun)")


@pytest.fixture()
def map_config():
    from gofr_amd.config import MapConfig
    return MapConfig({"APP_NAME": "test-app", "LOG_LEVEL": "FATAL"})


@pytest.fixture()
def app(map_config):
    """An App with a silent logger and no datasources, defaults installed."""
    import gofr_amd
    a = gofr_amd.New(config=map_config)
    a.install_default_routes()
    yield a
    a.shutdown()


def make_request(method="GET", path="/", query="", headers=None, body=b""):
    from gofr_amd.http.request import Request
    return Request(method=method, path=path, query_string=query,
                   headers=headers or {}, body=body)
