import asyncio
import inspect
import os

import pytest

# keep auth tests fast; production default stays 260k iterations
os.environ.setdefault("KAKVEDA_PBKDF2_ITERS", "1000")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an AMD GPU (MI355X) and the built HIP extension"
    )


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


# Minimal asyncio runner so async test functions work without pytest-asyncio.
@pytest.hookimpl(tryfirst=True)
def pytest_pyfunc_call(pyfuncitem):
    fn = pyfuncitem.obj
    if inspect.iscoroutinefunction(fn):
        kwargs = {
            name: pyfuncitem.funcargs[name]
            for name in pyfuncitem._fixtureinfo.argnames
        }
        asyncio.run(fn(**kwargs))
        return True
    return None
