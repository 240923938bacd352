import os
import sys
from pathlib import Path

import pytest

# repo root on sys.path so `import xotorch_amd` works without install
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

os.environ.setdefault("XOT_OFFLINE", "1")


def pytest_configure(config):
  config.addinivalue_line("markers", "gpu: needs a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
  import torch
  if torch.cuda.is_available():
    return
  skip_gpu = pytest.mark.skip(reason="no GPU in this container")
  for item in items:
    if "gpu" in item.keywords:
      item.add_marker(skip_gpu)
