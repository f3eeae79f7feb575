"""Test harness conventions.

- tests mirror the package one module per source module (unit/) plus
  integration lanes; fakes come from calfkit_amd.testing and
  calfkit_amd.adapters.memory, never test-local doubles
  (pattern: /root/reference/tests/unit/conftest.py:1-5,
   /root/reference/CONTRIBUTING.md:74-189).
- the ``gpu`` marker gates tests needing a real MI355X; on a box without a
  GPU they auto-skip, so a bare ``pytest tests`` is always green on CPU.
- async test functions run via asyncio.run (no pytest-asyncio in the image).
"""

import asyncio
import inspect

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run with -m gpu)")
    config.addinivalue_line("markers", "dist: multi-process distributed test")
    config.addinivalue_line("markers", "slow: long-running test")
    config.addinivalue_line(
        "markers",
        "kafka: needs a live broker (CALF_TEST_KAFKA_BOOTSTRAP)")


def _cuda_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def pytest_collection_modifyitems(config, items):
    gpu_marked = [i for i in items if i.get_closest_marker("gpu")]
    if gpu_marked and not _cuda_available():
        skip = pytest.mark.skip(reason="no GPU on this box")
        for item in gpu_marked:
            item.add_marker(skip)


@pytest.hookimpl(tryfirst=True)
def pytest_pyfunc_call(pyfuncitem):
    fn = pyfuncitem.obj
    if inspect.iscoroutinefunction(fn):
        kwargs = {name: pyfuncitem.funcargs[name]
                  for name in pyfuncitem._fixtureinfo.argnames}
        asyncio.run(fn(**kwargs))
        return True
    return None
