import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kata_xpu_device_plugin_amd.testing.mocknode import make_mock_node  # noqa: E402


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run via gpurun)")


@pytest.fixture
def mock_node(tmp_path):
    """Default 8×MI355X vfio-bound mock node (BASELINE.json config #1)."""
    return make_mock_node(str(tmp_path))


@pytest.fixture
def mock_cfg(mock_node):
    return mock_node.config()
