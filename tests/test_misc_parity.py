"""Small parity checks: conda wrapping, get_status, packaging."""

import asyncio
import subprocess
import sys
import tarfile
from pathlib import Path

from covalent_ssh_plugin_amd import SSHExecutor


def test_submit_command_plain():
    ex = SSHExecutor(username="u", hostname="h", python_path="python3")
    cmd = ex._submit_command(".cache/covalent/exec_d_0.py")
    assert cmd == "python3 .cache/covalent/exec_d_0.py"


def test_submit_command_conda_wrap():
    """conda activation wrapper parity (reference ssh.py:379-380)."""
    ex = SSHExecutor(username="u", hostname="h", conda_env="rocm-env")
    cmd = ex._submit_command("exec.py")
    assert cmd.startswith('eval "$(conda shell.bash hook)" && conda activate rocm-env && ')
    assert cmd.endswith("python exec.py")


def test_get_status_checks_file(local_executor, tmp_path):
    """get_status: ls-equality check parity (reference ssh.py:402-406)."""
    ex = local_executor()

    async def main():
        transport = await ex._client_connect()
        missing = await ex.get_status(transport, ".cache/covalent/nope.pkl")
        (local_executor.home / "present.pkl").write_bytes(b"x")
        present = await ex.get_status(transport, "present.pkl")
        return missing, present

    missing, present = asyncio.run(main())
    assert missing is False
    assert present is True


def test_sdist_builds(tmp_path):
    """Packaging integrity (reference CI builds + diffs the sdist,
    tests.yml): the sdist must build and contain the package, the HIP
    source, and both remote templates."""
    repo = Path(__file__).resolve().parent.parent
    out = subprocess.run(
        [sys.executable, "setup.py", "sdist", "--dist-dir", str(tmp_path)],
        cwd=repo,
        capture_output=True,
        text=True,
        timeout=180,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    (sdist,) = tmp_path.glob("*.tar.gz")
    with tarfile.open(sdist) as tf:
        names = tf.getnames()
    assert any(n.endswith("covalent_ssh_plugin_amd/ssh.py") for n in names)
    assert any(n.endswith("ops/hip/csp_gpu.hip") for n in names)
    assert any(n.endswith("remote/stub_template.py") for n in names)
    assert any(n.endswith("remote/worker_template.py") for n in names)


def test_executor_instance_is_picklable():
    """Covalent serializes executor instances into its dispatch records;
    the executor must survive a pickle round trip (transports/workers are
    module-level pooled state, not instance state)."""
    import pickle

    ex = SSHExecutor(username="u", hostname="h", gpu_slots=4, persistent_workers=True)
    ex2 = pickle.loads(pickle.dumps(ex))
    assert ex2.username == "u"
    assert ex2.hostname == "h"
    assert ex2.gpu_slots == 4
    assert ex2.persistent_workers is True


def test_pickled_executor_still_executes(tmp_path):
    """Covalent deserializes executor instances before calling run();
    a round-tripped instance must dispatch normally (pooled state is
    module-level and re-binds on first use)."""
    import asyncio
    import pickle

    home = tmp_path / "home"
    home.mkdir()
    ex = SSHExecutor(
        transport="local",
        local_home=str(home),
        cache_dir=str(tmp_path / "cache"),
        python_path=sys.executable,
    )
    ex2 = pickle.loads(pickle.dumps(ex))

    def fn(a, b):
        return a * b

    assert asyncio.run(ex2.execute(fn, [6, 7], {})) == 42


def test_ssh_extra_options_passthrough():
    ex = SSHExecutor(
        username="u",
        hostname="h",
        ssh_extra_options=["-o", "StrictHostKeyChecking=yes", "-J", "bastion"],
    )
    t = ex._make_transport()
    args = t._base_args()
    assert "StrictHostKeyChecking=yes" in args
    assert "-J" in args and "bastion" in args
    # different option sets must not share a pooled transport
    ex2 = SSHExecutor(username="u", hostname="h")
    assert ex._pool_key() != ex2._pool_key()
