import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parents[1]
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))

# In small shared containers torch's default all-core OMP pool thrashes:
# measured 6x SLOWER than single-thread on the suite's small CPU GEMMs.
try:
    import torch

    torch.set_num_threads(1)
except Exception:
    pass


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an AMD GPU (MI355X) to run")


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        have_gpu = torch.cuda.is_available()
    except Exception:
        have_gpu = False
    if have_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def sample_market():
    """Seeded random-walk OHLCV fixture (the reference's synthetic-dataframe
    idiom, tests/test_feature_window_preprocessor.py:12-21)."""
    from gymfx_amd.data.feed import synthetic_ohlcv

    return synthetic_ohlcv(600, seed=42, vol=3e-4, extra_feature_columns=3)
