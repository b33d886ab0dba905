import asyncio
import inspect
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

# The vault refuses to exist without key material (production parity with
# the reference's CLOAK_ENCRYPTION_KEY); the suite provides one like a
# deployment would.
os.environ.setdefault("QUORACLE_VAULT_KEY", "pytest-suite-vault-key")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)"
    )
    config.addinivalue_line(
        "markers", "asyncio: run the test function in a fresh asyncio event loop"
    )
    config.addinivalue_line(
        "markers", "gpu_experimental: unvalidated GPU kernels — run "
        "explicitly with -m gpu_experimental on a GPU box; excluded from "
        "the standard gpu suite"
    )


@pytest.hookimpl(tryfirst=True)
def pytest_pyfunc_call(pyfuncitem):
    """Minimal asyncio runner (pytest-asyncio is not in this image)."""
    func = pyfuncitem.obj
    if inspect.iscoroutinefunction(func):
        kwargs = {name: pyfuncitem.funcargs[name]
                  for name in pyfuncitem._fixtureinfo.argnames}
        asyncio.run(func(**kwargs))
        return True
    return None


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is present."""
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords or "gpu_experimental" in item.keywords:
            item.add_marker(skip_gpu)
