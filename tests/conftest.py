import sys
import os

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
  config.addinivalue_line(
      "markers", "gpu: test requires an MI355X GPU (run via gpurun)")
  config.addinivalue_line(
      "markers", "slow: multi-process / subprocess tests (still CPU-only)")


@pytest.fixture(autouse=True)
def _clear_gin():
  from tensor2robot_amd import gin
  gin.clear_config()
  yield
  gin.clear_config()
