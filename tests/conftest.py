import asyncio
import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
FIXTURES = os.path.join(os.path.dirname(os.path.abspath(__file__)), "fixtures")
for _p in (REPO_ROOT, FIXTURES):
    if _p not in sys.path:
        sys.path.insert(0, _p)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X GPU")
    config.addinivalue_line("markers", "asyncio: run test on a fresh asyncio loop")
    # make fixture handler modules importable in spawned runner containers
    existing = os.environ.get("PYTHONPATH", "")
    parts = [p for p in (REPO_ROOT, FIXTURES) if p not in existing.split(os.pathsep)]
    if parts:
        os.environ["PYTHONPATH"] = os.pathsep.join(parts + ([existing] if existing else []))


@pytest.fixture
def anyio_backend():
    return "asyncio"


@pytest.hookimpl(tryfirst=True)
def pytest_pyfunc_call(pyfuncitem):
    """Run async test functions on a fresh event loop (no pytest-asyncio in
    this image)."""
    fn = pyfuncitem.obj
    if asyncio.iscoroutinefunction(fn):
        kwargs = {
            name: pyfuncitem.funcargs[name]
            for name in pyfuncitem._fixtureinfo.argnames
        }
        loop = asyncio.new_event_loop()
        try:
            loop.run_until_complete(fn(**kwargs))
        finally:
            try:
                loop.run_until_complete(loop.shutdown_asyncgens())
            finally:
                loop.close()
        return True
    return None
