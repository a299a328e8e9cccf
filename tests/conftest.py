import asyncio
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run on the GPU box)")
    config.addinivalue_line(
        "markers", "functional_tests: live covalent server + real SSH target"
    )


@pytest.fixture(autouse=True)
def _fresh_pools():
    """Isolate module-level pooled state (transports, GPU slot tables)
    between tests."""
    from covalent_ssh_plugin_amd.gpu import slots
    from covalent_ssh_plugin_amd.remote import workers
    from covalent_ssh_plugin_amd.transport import pool

    pool.reset()
    slots.reset()
    workers.reset()
    yield
    # Close any workers/transports a test left running.
    try:
        asyncio.run(workers.close_all())
        asyncio.run(pool.close_all())
    except RuntimeError:
        pass
    slots.reset()


@pytest.fixture
def local_executor(tmp_path):
    """SSHExecutor over the loopback transport with isolated dirs."""
    from covalent_ssh_plugin_amd import SSHExecutor

    home = tmp_path / "remote_home"
    cache = tmp_path / "cache"
    home.mkdir()

    def make(**overrides):
        kwargs = dict(
            transport="local",
            local_home=str(home),
            cache_dir=str(cache),
            python_path=sys.executable,
        )
        kwargs.update(overrides)
        return SSHExecutor(**kwargs)

    make.home = home
    make.cache = cache
    return make


def run_async(coro):
    return asyncio.run(coro)
