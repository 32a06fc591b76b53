import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
if str(REPO) not in sys.path:
    sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    # Hard per-test cap on GPU tests: a wedged device call must surface as a
    # test failure, not consume the whole suite's wall clock. The "thread"
    # method fires even when the hang is inside a C call (it os._exit's).
    for item in items:
        if "gpu" in item.keywords and item.get_closest_marker("timeout") is None:
            item.add_marker(pytest.mark.timeout(300, method="thread"))
