# SPDX-License-Identifier: Apache-2.0
import os
import sys

import pytest

os.environ.setdefault("OMP_WAIT_POLICY", "PASSIVE")

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def _tame_cpu_threads():
    import torch
    if not torch.cuda.is_available():
        # cgroup-throttled CPU containers: spinning thread pools are
        # counterproductive (measured 1 thread 1.8ms vs 4 threads 11ms
        # for the same elementwise op)
        torch.set_num_threads(1)


_tame_cpu_threads()


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session", autouse=True)
def _ensure_cpu_ext():
    """Build the CPU extension if missing (seconds with g++)."""
    from legate_sparse import _cext
    if not _cext.has_cpu():
        from legate_sparse._build import build_cpu
        build_cpu()
        import importlib
        import legate_sparse._cext as m
        importlib.reload(m)
    yield


@pytest.fixture
def rng():
    import numpy as np
    return np.random.default_rng(42)
