import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
if str(REPO) not in sys.path:
    sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


def pytest_collection_modifyitems(config, items):
    try:
        import ps_lite_amd  # noqa

        has_gpu = ps_lite_amd.gpu_count() > 0
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU on this machine")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
