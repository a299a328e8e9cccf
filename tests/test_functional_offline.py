"""The functional-test tier, executed OFFLINE ("the
functional/covalent tier has never executed").

Mirrors tests/functional_tests/* (reference analogs:
tests/functional_tests/basic_workflow_test.py, svm_workflow.py) but runs
here, in CI, with no live server and no sshd:

* the covalent host is tests/fake_covalent — its workflow engine builds
  the lattice DAG and resolves ``executor="ssh"`` through the REAL
  setuptools entry point, instantiating the plugin config-driven,
  exactly like covalent's plugin loader;
* the SSH hop is the PATH-shim ssh client, so electrons travel the full
  OpenSSH-transport pipeline (tar staging, sentinel-framed results).

Each test runs in a subprocess so compat.py binds to the fake covalent
package at import time (PYTHONPATH), the same wiring a real install has.
The live-server tier in tests/functional_tests/ remains marker-gated
for when a real covalent + SSH target exists.
"""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
FAKE = REPO / "tests" / "fake_covalent"
SHIM = REPO / "tests" / "sshim"

_PRELUDE = """
import json, sys
import covalent as ct
from covalent._shared_files.config import set_config

set_config({
    "executors.ssh.username": "mi355x",
    "executors.ssh.hostname": "functional-node.test",
    "executors.ssh.ssh_key_file": __import__("os").environ["FAKE_KEY"],
    "executors.ssh.cache_dir": __import__("os").environ["FAKE_CACHE"],
    "executors.ssh.python_path": sys.executable,
    "executors.ssh.transport": "ssh",
})
"""


def run_workflow(code: str, sshim, tmp_path) -> dict:
    key = tmp_path / "id_func"
    key.write_text("fake key\n")
    cache = tmp_path / "func_cache"
    env = dict(os.environ)
    env["PYTHONPATH"] = f"{FAKE}:{REPO}"
    env["PATH"] = f"{SHIM}{os.pathsep}{env.get('PATH', '')}"
    env["SSHIM_HOME"] = str(sshim.home)
    env["FAKE_KEY"] = str(key)
    env["FAKE_CACHE"] = str(cache)
    proc = subprocess.run(
        [sys.executable, "-c", _PRELUDE + code],
        capture_output=True,
        text=True,
        timeout=240,
        env=env,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    return json.loads(proc.stdout.splitlines()[-1])


def test_basic_workflow_completes_offline(sshim, tmp_path):
    out = run_workflow(
        """
@ct.electron(executor="ssh")
def join_words(a, b):
    return ", ".join([a, b])

@ct.electron
def excitement(phrase):
    return f"{phrase}!"

@ct.lattice
def simple_workflow(a, b):
    phrase = join_words(a, b)
    return excitement(phrase)

dispatch_id = ct.dispatch(simple_workflow)("Hello", "World")
result = ct.get_result(dispatch_id, wait=True)
print(json.dumps({"status": str(result.status), "result": result.result}))
""",
        sshim,
        tmp_path,
    )
    assert out == {"status": "COMPLETED", "result": "Hello, World!"}


def test_failing_electron_fails_lattice_offline(sshim, tmp_path):
    out = run_workflow(
        """
@ct.electron(executor="ssh")
def boom():
    raise RuntimeError("intentional failure")

@ct.lattice
def failing_workflow():
    return boom()

dispatch_id = ct.dispatch(failing_workflow)()
result = ct.get_result(dispatch_id, wait=True)
print(json.dumps({"status": str(result.status), "error": result.error}))
""",
        sshim,
        tmp_path,
    )
    assert out["status"] == "FAILED"
    assert "intentional failure" in out["error"]


def test_svm_workflow_offline(sshim, tmp_path):
    """3-electron ML pipeline, only training over SSH, DepsPip-guarded,
    non-trivial pickled model return (reference svm_workflow.py)."""
    import pytest

    pytest.importorskip("sklearn")
    out = run_workflow(
        """
import numpy as np

deps = ct.DepsPip(packages=["scikit-learn"])

@ct.electron
def make_data(n=200):
    rng = np.random.default_rng(0)
    x = rng.normal(size=(n, 4))
    y = (x[:, 0] + x[:, 1] > 0).astype(int)
    return x, y

@ct.electron(executor="ssh", deps_pip=deps)
def train_svm(data):
    from sklearn.svm import SVC

    x, y = data
    return SVC(kernel="linear").fit(x, y)

@ct.electron
def score(clf, data):
    x, y = data
    return clf.score(x, y)

@ct.lattice
def workflow():
    data = make_data()
    clf = train_svm(data)
    return score(clf, data)

dispatch_id = ct.dispatch(workflow)()
result = ct.get_result(dispatch_id, wait=True)
print(json.dumps({"status": str(result.status), "score": result.result}))
""",
        sshim,
        tmp_path,
    )
    assert out["status"] == "COMPLETED"
    assert out["score"] > 0.8


def test_deps_pip_missing_package_fails_offline(sshim, tmp_path):
    """DepsPip points at a package absent on the remote: the electron
    (and lattice) must FAIL, not silently skip the dependency."""
    out = run_workflow(
        """
deps = ct.DepsPip(packages=["definitely-not-a-real-package-xyz"])

@ct.electron(executor="ssh", deps_pip=deps)
def needs_missing():
    return 1

@ct.lattice
def wf():
    return needs_missing()

dispatch_id = ct.dispatch(wf)()
result = ct.get_result(dispatch_id, wait=True)
print(json.dumps({"status": str(result.status), "error": result.error}))
""",
        sshim,
        tmp_path,
    )
    assert out["status"] == "FAILED"


def test_gpu_slot_workflow_offline(sshim, tmp_path):
    """The gpu-identity workflow with a simulated GPU endpoint: the
    electron must see its CSP_GPU_SLOT -> HIP_VISIBLE_DEVICES pinning
    through the whole covalent->plugin->shim-SSH stack."""
    out = run_workflow(
        """
from covalent import _resolve_executor
from covalent_ssh_plugin_amd.transport import pool as transport_pool

probe_ex = _resolve_executor("ssh")
transport_pool.store_check(probe_ex._pool_key(), "env", (True, "", "", True))

@ct.electron(executor="ssh")
def gpu_identity():
    import os

    return {
        "slot": os.environ.get("CSP_GPU_SLOT"),
        "visible": os.environ.get("HIP_VISIBLE_DEVICES"),
    }

@ct.lattice
def probe_workflow():
    return gpu_identity()

dispatch_id = ct.dispatch(probe_workflow)()
result = ct.get_result(dispatch_id, wait=True)
print(json.dumps({"status": str(result.status), "result": result.result}))
""",
        sshim,
        tmp_path,
    )
    assert out["status"] == "COMPLETED"
    assert out["result"]["slot"] is not None
    assert out["result"]["visible"] == out["result"]["slot"]
