"""Remote stub contract tests: render the template, run it standalone
(the way sshd+python would), verify the on-disk formats of SURVEY.md
§2.3."""

import pickle
import subprocess
import sys
from pathlib import Path

import cloudpickle
import pytest

from covalent_ssh_plugin_amd.remote.stub import render_stub


def _run_stub(tmp_path, fn, args=(), kwargs=None, workdir=None, env=None):
    function_file = tmp_path / "function_t_0.pkl"
    result_file = tmp_path / "result_t_0.pkl"
    meta_file = tmp_path / "meta_t_0.json"
    script_file = tmp_path / "exec_t_0.py"
    with open(function_file, "wb") as f:
        cloudpickle.dump((fn, list(args), dict(kwargs or {})), f)
    script = render_stub(
        remote_result_file=str(result_file),
        remote_function_file=str(function_file),
        current_remote_workdir=str(workdir or tmp_path / "wd"),
        remote_meta_file=str(meta_file),
    )
    script_file.write_text(script)
    proc = subprocess.run(
        [sys.executable, str(script_file)],
        capture_output=True,
        cwd=tmp_path,
        env=env,
        timeout=60,
    )
    return proc, result_file, meta_file


def test_render_has_no_placeholders(tmp_path):
    script = render_stub(
        remote_result_file="r.pkl",
        remote_function_file="f.pkl",
        current_remote_workdir="wd",
    )
    assert "__CSP_" not in script
    compile(script, "exec_stub.py", "exec")  # syntactically valid


def test_success_roundtrip(tmp_path):
    proc, result_file, meta_file = _run_stub(tmp_path, lambda x, y: x * y, (6, 7))
    assert proc.returncode == 0, proc.stderr
    result, exception = pickle.loads(result_file.read_bytes())
    assert result == 42 and exception is None
    assert meta_file.exists()


def test_exception_in_pickle_exit_zero(tmp_path):
    """Task exceptions travel in the pickle; exit code stays 0
    (reference exec.py:37-46)."""

    def boom():
        raise KeyError("k")

    proc, result_file, _ = _run_stub(tmp_path, boom)
    assert proc.returncode == 0
    result, exception = pickle.loads(result_file.read_bytes())
    assert result is None and isinstance(exception, KeyError)


def test_workdir_created_and_chdir(tmp_path):
    wd = tmp_path / "deep" / "workdir"

    def get_cwd():
        import os

        return os.getcwd()

    proc, result_file, _ = _run_stub(tmp_path, get_cwd, workdir=wd)
    assert proc.returncode == 0
    result, exception = pickle.loads(result_file.read_bytes())
    assert exception is None
    assert Path(result) == wd
    assert wd.is_dir()


def test_namedtuple_result(tmp_path):
    def make_nt():
        import collections

        Point = collections.namedtuple("Point", "x y")
        return Point(1, 2)

    proc, result_file, _ = _run_stub(tmp_path, make_nt)
    result, exception = pickle.loads(result_file.read_bytes())
    assert exception is None
    assert tuple(result) == (1, 2)


def test_missing_gpu_lib_with_slot_is_loud(tmp_path):
    """A GPU slot assignment + a broken/missing HIP library must surface
    an error in the result pickle, never a silent CPU run."""
    import os

    function_file = tmp_path / "f.pkl"
    result_file = tmp_path / "r.pkl"
    script_file = tmp_path / "e.py"
    with open(function_file, "wb") as f:
        cloudpickle.dump((lambda: 1, [], {}), f)
    script = render_stub(
        remote_result_file=str(result_file),
        remote_function_file=str(function_file),
        current_remote_workdir=str(tmp_path / "wd"),
        gpu_lib_path=str(tmp_path / "does_not_exist.so"),
    )
    script_file.write_text(script)
    env = dict(os.environ)
    env["CSP_GPU_SLOT"] = "0"
    proc = subprocess.run(
        [sys.executable, str(script_file)], capture_output=True, cwd=tmp_path, env=env, timeout=60
    )
    assert proc.returncode == 0
    result, exception = pickle.loads(result_file.read_bytes())
    assert result is None
    assert exception is not None  # OSError from ctypes.CDLL


def test_gpu_slot_resolves_within_ambient_visibility(tmp_path):
    """CSP_GPU_SLOT must select WITHIN an ambient HIP_VISIBLE_DEVICES
    list (pod GPU isolation), not clobber it."""
    import os

    def visible():
        import os

        return os.environ.get("HIP_VISIBLE_DEVICES")

    env = dict(os.environ)
    env["HIP_VISIBLE_DEVICES"] = "5,7"
    env["CSP_GPU_SLOT"] = "1"
    proc, result_file, _ = _run_stub(tmp_path, visible, env=env)
    assert proc.returncode == 0
    result, exception = pickle.loads(result_file.read_bytes())
    assert exception is None
    assert result == "7"

    env["CSP_GPU_SLOT"] = "0"
    proc, result_file, _ = _run_stub(tmp_path, visible, env=env)
    result, _ = pickle.loads(result_file.read_bytes())
    assert result == "5"

    # no ambient restriction -> slot index used directly
    env.pop("HIP_VISIBLE_DEVICES")
    env["CSP_GPU_SLOT"] = "3"
    proc, result_file, _ = _run_stub(tmp_path, visible, env=env)
    result, _ = pickle.loads(result_file.read_bytes())
    assert result == "3"


def test_unpicklable_result_in_pickle_not_crash(tmp_path):
    """An unpicklable result becomes (None, TypeError) in the result
    pickle with exit 0 (the worker path behaves the same)."""

    def gen():
        return (i for i in range(3))

    proc, result_file, _ = _run_stub(tmp_path, gen)
    assert proc.returncode == 0, proc.stderr
    result, exception = pickle.loads(result_file.read_bytes())
    assert result is None and isinstance(exception, TypeError)
