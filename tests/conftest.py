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


_EXIT_STATUS = {"status": 0}


def pytest_sessionfinish(session, exitstatus):
    _EXIT_STATUS["status"] = int(exitstatus)


def pytest_unconfigure(config):
    """Exit without CPython finalization: daemon event-loop threads (DHT loops
    shared across tests) can touch the interpreter during Py_Finalize and
    SIGABRT/SEGV AFTER all tests passed (~1/3 of full-suite runs), turning a
    green run into rc=134/139. This runs after the terminal summary is
    printed; stop the loops we can find, flush, then _exit with pytest's own
    status."""
    import threading

    try:
        from hivemind_amd.utils.asyncio_utils import EventLoopThread

        for thread in threading.enumerate():
            if isinstance(thread, EventLoopThread):
                try:
                    thread.shutdown(timeout=1.0)
                except Exception:
                    pass
    except Exception:
        pass
    sys.stdout.flush()
    sys.stderr.flush()
    os._exit(_EXIT_STATUS["status"])


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
