import asyncio
import os
import sys

import pytest

# Make the repo root importable regardless of pytest invocation directory.
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires a real MI355X GPU (run on the GPU box)")


@pytest.fixture
def run():
    """Run a coroutine to completion (we avoid a pytest-asyncio dependency)."""

    def _run(coro):
        return asyncio.run(coro)

    return _run


def require_gpu():
    try:
        import torch

        if not torch.cuda.is_available():
            pytest.skip("no GPU available")
    except Exception:
        pytest.skip("torch unavailable")
