"""Exercise the COVALENT_AVAILABLE=True wiring of compat.py using a
minimal fake covalent package on PYTHONPATH (the real covalent cannot be
installed offline).  Runs in a subprocess because compat.py resolves the
import once at module load."""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
FAKE = Path(__file__).resolve().parent / "fake_covalent"


def run_with_fake_covalent(code: str) -> dict:
    env = dict(os.environ)
    env["PYTHONPATH"] = f"{FAKE}:{REPO}"
    proc = subprocess.run(
        [sys.executable, "-c", code], capture_output=True, text=True, timeout=120, env=env
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    return json.loads(proc.stdout.splitlines()[-1])


def test_compat_binds_to_covalent():
    out = run_with_fake_covalent(
        """
import json
from covalent_ssh_plugin_amd import compat
from covalent.executor.executor_plugins.remote_executor import RemoteExecutor
print(json.dumps({
    "available": compat.COVALENT_AVAILABLE,
    "base_is_covalent": compat.RemoteExecutor is RemoteExecutor,
}))
"""
    )
    assert out == {"available": True, "base_is_covalent": True}


def test_executor_reads_covalent_config():
    """Ctor resolution must consult covalent's get_config (arg -> config
    -> default) when covalent is present."""
    out = run_with_fake_covalent(
        """
import json
from covalent_ssh_plugin_amd import SSHExecutor
ex = SSHExecutor()  # no args: everything from covalent config/defaults
ex2 = SSHExecutor(username="explicit")
print(json.dumps({
    "username": ex.username,
    "hostname": ex.hostname,
    "python_path": ex.python_path,
    "remote_cache": ex.remote_cache,
    "explicit_wins": ex2.username,
    "is_remote_executor_subclass": type(ex).__mro__[1].__module__,
}))
"""
    )
    assert out["username"] == "cfg-user"
    assert out["hostname"] == "cfg-host"
    assert out["python_path"] == "python3"
    assert out["remote_cache"] == ".cache/covalent"  # hardened default
    assert out["explicit_wins"] == "explicit"
    assert "remote_executor" in out["is_remote_executor_subclass"]


def test_dispatch_works_under_covalent_base(tmp_path):
    """Full electron round trip with the covalent base class in the MRO."""
    out = run_with_fake_covalent(
        f"""
import asyncio, json, sys, tempfile
from covalent_ssh_plugin_amd import SSHExecutor

async def main():
    with tempfile.TemporaryDirectory() as home, tempfile.TemporaryDirectory() as cache:
        ex = SSHExecutor(transport="local", local_home=home, cache_dir=cache,
                         python_path=sys.executable)
        r = await ex.run(lambda a, b: a + b, [20, 22], {{}},
                         {{"dispatch_id": "cv", "node_id": 0}})
        await SSHExecutor.close_pool()
        return r

print(json.dumps({{"result": asyncio.run(main())}}))
"""
    )
    assert out == {"result": 42}
