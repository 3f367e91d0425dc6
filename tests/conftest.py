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


def pytest_sessionfinish(session, exitstatus):
    """Exit without CPython finalization: daemon event-loop threads (DHT loops
    shared across tests) can touch the interpreter during Py_Finalize and
    SIGABRT AFTER all tests passed (~1/3 of full-suite runs), turning a green
    run into rc=134. Stop the loops we can find, flush the report, then
    _exit with pytest's own status."""
    import threading

    from hivemind_amd.utils.asyncio_utils import EventLoopThread

    for thread in threading.enumerate():
        if isinstance(thread, EventLoopThread):
            try:
                thread.shutdown(timeout=1.0)
            except Exception:
                pass
    sys.stdout.flush()
    sys.stderr.flush()
    os._exit(int(exitstatus))


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
