import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    try:
        from scanner_amd import _core
        have = _core.have_gpu()
    except Exception:
        have = False
    if have:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def sc(tmp_path):
    import scanner_amd as sp
    return sp.Client(db_path=str(tmp_path / "db"))


def make_smooth_video(n=30, h=48, w=64):
    """Moving-gradient clip: temporally smooth, realistic for codec tests."""
    yy, xx = np.mgrid[0:h, 0:w]
    frames = np.zeros((n, h, w, 3), np.uint8)
    for i in range(n):
        frames[i, :, :, 0] = (xx + i * 2) % 256
        frames[i, :, :, 1] = (yy + i) % 256
        frames[i, :, :, 2] = (xx + yy + i * 3) % 256
    return frames


def make_video(n=30, h=48, w=64, c=3, seed=0):
    """Synthetic moving-gradient clip (deterministic)."""
    rng = np.random.RandomState(seed)
    base = rng.randint(0, 255, size=(h, w, c)).astype(np.uint8)
    frames = np.zeros((n, h, w, c), dtype=np.uint8)
    for i in range(n):
        frames[i] = np.roll(base, i, axis=1)
        frames[i, :, :, 0] = (frames[i, :, :, 0].astype(np.int32)
                              + i) % 256
    return frames
