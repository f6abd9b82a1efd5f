import os
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if ROOT not in sys.path:
    sys.path.insert(0, ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X)")
    # Build the native libraries if they are missing (hipcc cross-compiles
    # for gfx950 without a GPU; the oracle is plain g++).
    from tools.build import build_all
    build_all()


@pytest.fixture(scope="session")
def engine():
    """Product engine on hip device 0 (gpu tests only)."""
    import tikv_amd
    eng = tikv_amd.Engine(0)
    yield eng
    eng.close()
