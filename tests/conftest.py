import asyncio
import inspect

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")
    config.addinivalue_line(
        "markers", "asyncio: coroutine test (run via asyncio.run)")


def pytest_pyfunc_call(pyfuncitem):
    """Minimal async test support (pytest-asyncio is not in this image)."""
    fn = pyfuncitem.obj
    if inspect.iscoroutinefunction(fn):
        kwargs = {n: pyfuncitem.funcargs[n]
                  for n in pyfuncitem._fixtureinfo.argnames}
        asyncio.run(asyncio.wait_for(fn(**kwargs), timeout=60))
        return True
    return None


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def pytest_runtest_logreport(report):
    """Append every failure to .pytest_failures.log — flaky failures in
    CI loops otherwise vanish with the scrollback."""
    if report.failed and report.when == "call":
        import time as _t
        with open(".pytest_failures.log", "a") as f:
            f.write(f"{_t.strftime('%H:%M:%S')} {report.nodeid}\n")
