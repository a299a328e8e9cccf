"""Full executor pipeline over the REAL OpenSSH transport via the
PATH-shim ssh client.

Everything the loopback suite exercises runs here through actual `ssh`
argv: fused tar-on-stdin dispatch, the discrete template path,
persistent-worker channels (including out-of-band tensor frames), the
connect-retry matrix, failure policy, workdirs and cleanup.  Reference
flow being reproduced: /root/reference/covalent_ssh_plugin/ssh.py:466-591.
"""

import asyncio
import os
import sys

import pytest

from covalent_ssh_plugin_amd import SSHExecutor


def _make_add():
    def add(x, y):
        return x + y

    return add


_add = _make_add()


def test_fused_roundtrip_over_ssh(sshim_executor):
    ex = sshim_executor()
    result = asyncio.run(ex.execute(_add, [3, 4], {}, dispatch_id="d", node_id=0))
    assert result == 7
    rec = ex.last_task_record
    assert rec.operation_id == "d_0"
    assert len(rec.remote_meta["result_sha256"]) == 64
    # remote task files were cleaned up inside the same round trip
    rc = sshim_executor.home / ".cache" / "covalent"
    leftovers = [p.name for p in rc.glob("*d_0*")]
    assert leftovers == []


def test_template_path_over_ssh(sshim_executor):
    """Discrete upload/submit/poll/fetch/cleanup round trips (reference
    §3.1 flow) through the real client."""
    ex = sshim_executor(batch_roundtrips=False, poll_freq=1)
    result = asyncio.run(ex.execute(_add, [1], {"y": 2}, dispatch_id="d", node_id=1))
    assert result == 3
    assert "upload" in ex.last_task_record.phases
    assert "poll" in ex.last_task_record.phases
    rc = sshim_executor.home / ".cache" / "covalent"
    assert [p.name for p in rc.glob("*d_1*")] == []


def test_exception_roundtrip_over_ssh(sshim_executor):
    def boom():
        raise ValueError("inner ssh failure")

    ex = sshim_executor()
    with pytest.raises(ValueError, match="inner ssh failure"):
        asyncio.run(ex.execute(boom, [], {}))


def test_workdirs_over_ssh(sshim_executor):
    def cwd_name():
        import os

        return os.getcwd()

    ex = sshim_executor(create_unique_workdir=True)
    out = asyncio.run(ex.execute(cwd_name, [], {}, dispatch_id="dx", node_id=5))
    assert out == str(sshim_executor.home / "covalent-workdir" / "dx" / "node_5")


def test_task_stdout_noise_over_ssh(sshim_executor):
    def chatty():
        print("noise \x00\xff --CSP-RESULT-- more noise")
        sys.stderr.write("stderr noise\n")
        return 42

    ex = sshim_executor()
    assert asyncio.run(ex.execute(chatty, [], {})) == 42


def test_worker_mode_over_ssh_reuses_process(sshim_executor):
    """Persistent workers ride ONE long-lived ssh channel; the second
    electron reuses the same remote pid."""
    ex = sshim_executor(persistent_workers=True)

    async def go():
        try:
            r1 = await ex.execute(_add, [1, 1], {}, dispatch_id="w", node_id=0)
            m1 = dict(ex.last_task_record.remote_meta)
            r2 = await ex.execute(_add, [2, 2], {}, dispatch_id="w", node_id=1)
            m2 = dict(ex.last_task_record.remote_meta)
            return r1, m1, r2, m2
        finally:
            await SSHExecutor.close_pool()

    r1, m1, r2, m2 = asyncio.run(go())
    assert (r1, r2) == (2, 4)
    assert m1["pid"] == m2["pid"]
    assert m2["served"] == m1["served"] + 1


def test_worker_oob_tensor_frames_over_ssh(sshim_executor):
    """Large tensor results travel as raw out-of-band frames through the
    shim client byte-exact (the worker protocol's staging path)."""
    torch = pytest.importorskip("torch")

    def make_tensor(n):
        import torch

        return torch.arange(n, dtype=torch.float32)

    ex = sshim_executor(
        persistent_workers=True, pinned_staging_threshold_bytes=1 << 20
    )

    async def go():
        try:
            return await ex.execute(make_tensor, [1 << 21], {})  # 8 MiB
        finally:
            await SSHExecutor.close_pool()

    out = asyncio.run(go())
    assert out.shape == (1 << 21,)
    assert out[-1].item() == float((1 << 21) - 1)
    assert ex.last_task_record.remote_meta["staging"]["mode"] == "cpu-oob"


def test_concurrent_fan_over_ssh(sshim_executor):
    """A fan of concurrent electrons multiplexed over the shim client."""
    ex = sshim_executor(persistent_workers=True, cpu_workers=4)

    def square(i):
        return i * i

    async def go():
        try:
            return await asyncio.gather(
                *[
                    ex.execute(square, [i], {}, dispatch_id="fan", node_id=i)
                    for i in range(16)
                ]
            )
        finally:
            await SSHExecutor.close_pool()

    results = asyncio.run(go())
    assert results == [i * i for i in range(16)]


def test_connect_retry_matrix_over_ssh(sshim_executor, sshim):
    """Reference retry policy (reference ssh.py:237-282) end to end:
    N scripted refusals then success."""
    fail = sshim.tmp / "failctr"
    fail.write_text("2")
    os.environ["SSHIM_FAIL_FILE"] = str(fail)
    try:
        ex = sshim_executor(max_connection_attempts=4, retry_wait_time=0)
        assert asyncio.run(ex.execute(_add, [5, 6], {})) == 11
        assert fail.read_text().strip() == "0"
    finally:
        os.environ.pop("SSHIM_FAIL_FILE", None)


def test_connect_exhaustion_raises_over_ssh(sshim_executor, sshim):
    fail = sshim.tmp / "failctr"
    fail.write_text("99")
    os.environ["SSHIM_FAIL_FILE"] = str(fail)
    try:
        ex = sshim_executor(max_connection_attempts=2, retry_wait_time=0)
        with pytest.raises(RuntimeError, match="[Cc]ould not connect"):
            asyncio.run(ex.execute(_add, [1, 2], {}))
    finally:
        os.environ.pop("SSHIM_FAIL_FILE", None)


def test_connect_failure_local_fallback_over_ssh(sshim_executor, sshim):
    fail = sshim.tmp / "failctr"
    fail.write_text("99")
    os.environ["SSHIM_FAIL_FILE"] = str(fail)
    try:
        ex = sshim_executor(
            max_connection_attempts=2, retry_wait_time=0, run_local_on_ssh_fail=True
        )
        # reference behavior: compute on the dispatcher (reference
        # ssh.py:202-204)
        assert asyncio.run(ex.execute(_add, [5, 20], {})) == 25
    finally:
        os.environ.pop("SSHIM_FAIL_FILE", None)


def test_remote_python_missing_over_ssh(sshim_executor):
    ex = sshim_executor(python_path="/nonexistent/python3")
    with pytest.raises(RuntimeError):
        asyncio.run(ex.execute(_add, [1, 2], {}))


def _with_fake_conda(monkeypatch):
    import os
    from pathlib import Path

    fakebin = Path(__file__).parent / "sshim" / "fakebin"
    monkeypatch.setenv(
        "SSHIM_REMOTE_PATH", f"{fakebin}{os.pathsep}{os.environ.get('PATH', '')}"
    )


def test_conda_env_activation_end_to_end(sshim_executor, monkeypatch):
    """The reference's conda wrapper (reference ssh.py:379-380) through
    the real client: hook eval'd, env verified, task runs activated."""
    _with_fake_conda(monkeypatch)
    ex = sshim_executor(conda_env="ml")

    def which_env():
        import os

        return os.environ.get("CONDA_DEFAULT_ENV")

    assert asyncio.run(ex.execute(which_env, [], {})) == "ml"


def test_conda_env_missing_fails(sshim_executor, monkeypatch):
    """A conda_env absent from `conda env list` fails the env check
    (reference ssh.py:508-519 behavior)."""
    _with_fake_conda(monkeypatch)
    ex = sshim_executor(conda_env="nonexistent-env")
    with pytest.raises(RuntimeError, match="not found"):
        asyncio.run(ex.execute(_add, [1, 2], {}))


def test_conda_env_worker_mode(sshim_executor, monkeypatch):
    """Persistent workers launch inside the activated env too."""
    _with_fake_conda(monkeypatch)
    ex = sshim_executor(conda_env="ml", persistent_workers=True)

    def which_env():
        import os

        return os.environ.get("CONDA_DEFAULT_ENV")

    async def go():
        try:
            return await ex.execute(which_env, [], {})
        finally:
            await SSHExecutor.close_pool()

    assert asyncio.run(go()) == "ml"
