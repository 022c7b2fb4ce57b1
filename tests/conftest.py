import sys
from pathlib import Path

import pytest

# repo root on sys.path so `vantage6_amd` and `algorithms` import without install
ROOT = Path(__file__).resolve().parent.parent
if str(ROOT) not in sys.path:
    sys.path.insert(0, str(ROOT))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU (run via gpurun)")
