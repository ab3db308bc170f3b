import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)"
    )


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU on this machine")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def small_market():
    """Deterministic small synthetic market shared across tests."""
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    ohlcv = generate_ohlcv(3000, 4, seed=7)
    return candles_chl_v(ohlcv)
