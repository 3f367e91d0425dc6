import asyncio
import gc
import os
import sys

import psutil
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")
    config.addinivalue_line("markers", "forked: legacy marker accepted for compatibility")


@pytest.fixture(autouse=True)
def cleanup_children():
    yield
    gc.collect()
    children = psutil.Process().children(recursive=True)
    if children:
        gone, alive = psutil.wait_procs(children, timeout=1)
        for child in alive:
            try:
                child.terminate()
            except psutil.NoSuchProcess:
                pass
        gone, alive = psutil.wait_procs(alive, timeout=3)
        for child in alive:
            try:
                child.kill()
            except psutil.NoSuchProcess:
                pass


@pytest.fixture
def event_loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()
