import asyncio
import inspect

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X)")
    config.addinivalue_line("markers", "slow: long-running tests")


@pytest.hookimpl(tryfirst=True)
def pytest_pyfunc_call(pyfuncitem):
    """Run ``async def`` tests without pytest-asyncio."""
    fn = pyfuncitem.obj
    if inspect.iscoroutinefunction(fn):
        kwargs = {
            name: pyfuncitem.funcargs[name]
            for name in pyfuncitem._fixtureinfo.argnames
        }
        asyncio.run(fn(**kwargs))
        return True
    return None
