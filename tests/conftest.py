"""Test harness config.

* registers the ``gpu`` marker (tests needing an MI355X; skipped on hosts
  without a HIP device)
* provides a minimal async-test runner (pytest-asyncio is not available in
  this image): any ``async def`` test is executed under ``asyncio.run`` with
  a per-test timeout.
"""
from __future__ import annotations

import os

# ROCm multiplexes streams onto GPU_MAX_HW_QUEUES hardware queues
# (default 4); oversubscription time-slices co-mapped streams at ~ms
# granularity. Must be set before the FIRST HIP init in the process
# (torch's or ours) — see ROUND2_NOTES.md "hardware-queue starvation".
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import asyncio
import inspect
import os
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))

ASYNC_TEST_TIMEOUT = float(os.environ.get("STARWAY_TEST_TIMEOUT", "180"))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD Instinct GPU (MI355X)"
    )


def pytest_collection_modifyitems(config, items):
    try:
        import starway_amd

        has_gpu = starway_amd.gpu_available()
    except Exception:
        has_gpu = False
    skip_gpu = pytest.mark.skip(reason="no HIP device visible on this host")
    for item in items:
        if "gpu" in item.keywords and not has_gpu:
            item.add_marker(skip_gpu)


@pytest.hookimpl(tryfirst=True)
def pytest_pyfunc_call(pyfuncitem):
    fn = pyfuncitem.obj
    if inspect.iscoroutinefunction(fn):
        kwargs = {
            name: pyfuncitem.funcargs[name]
            for name in pyfuncitem._fixtureinfo.argnames
        }
        asyncio.run(asyncio.wait_for(fn(**kwargs), timeout=ASYNC_TEST_TIMEOUT))
        return True
    return None


@pytest.fixture
def port():
    """A free TCP port from the kernel's ephemeral range (random fixed-range
    picks collided intermittently across the suite). Our listener sets
    SO_REUSEADDR, so the close->rebind window is safe."""
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p
