"""OpenSSHTransport exercised end to end through the PATH-shim ssh client.

Round 1 faked ``OpenSSHTransport._ssh_exec`` in every test, leaving the
flagship transport unverified ("What's missing" #1).  Here
the REAL transport spawns the REAL client argv; only the binary on PATH
is the shim (tests/sshim/ssh), which executes commands the way sshd
would (bash -c against a fake home, sanitized env).  Reference role:
asyncssh.connect/scp at /root/reference/covalent_ssh_plugin/ssh.py:263-268,
360-361, 451.
"""

import asyncio
import os

import pytest

from covalent_ssh_plugin_amd.transport import (
    OpenSSHTransport,
    TransportConnectError,
)


def make_transport(sshim, **kw):
    kwargs = dict(
        hostname=sshim.hostname,
        username="mi355x",
        ssh_key_file=str(sshim.key),
        control_dir=str(sshim.tmp / "ctl"),
    )
    kwargs.update(kw)
    return OpenSSHTransport(**kwargs)


def test_connect_and_run(sshim):
    async def go():
        t = make_transport(sshim)
        await t.connect()
        assert t.is_connected
        proc = await t.run("echo hello && pwd")
        assert proc.ok
        out = proc.text_out().splitlines()
        assert out[0] == "hello"
        assert out[1] == str(sshim.home)
        await t.close()

    asyncio.run(go())


def test_client_argv_construction(sshim):
    """The argv the transport hands the ssh client carries -p/-i/-l and
    terminates options with -- before the command."""

    async def go():
        t = make_transport(sshim, port=2222, extra_options=["-o", "Compression=yes"])
        await t.connect()
        await t.run("true")
        await t.close()

    asyncio.run(go())
    calls = sshim.log.read_text().splitlines()
    run_call = [c for c in calls if " -- true" in c or c.endswith("-- true")][0]
    assert "-p 2222" in run_call
    assert f"-i {sshim.key}" in run_call
    assert "-l mi355x" in run_call
    assert "Compression=yes" in run_call
    assert "BatchMode=yes" in run_call
    assert f"{sshim.hostname} -- " in run_call


def test_connect_refused_raises(sshim):
    async def go():
        t = make_transport(sshim, hostname="unreachable.invalid")
        with pytest.raises(TransportConnectError):
            await t.connect()
        assert not t.is_connected

    asyncio.run(go())


def test_remote_nonzero_exit_passes_through(sshim):
    """A remote command's own failure (< 255) is NOT a transport error —
    it surfaces through CompletedCommand.returncode (reference analog:
    SSHCompletedProcess.exit_status, reference ssh.py:383, 553)."""

    async def go():
        t = make_transport(sshim)
        await t.connect()
        proc = await t.run("echo oops >&2; exit 3")
        assert proc.returncode == 3
        assert "oops" in proc.text_err()
        assert t.is_connected
        await t.close()

    asyncio.run(go())


def test_rc255_marks_disconnected(sshim):
    """rc 255 from the client means the channel died: the transport
    surfaces TransportConnectError and flags itself disconnected so the
    pool reconnects (then a later connect() succeeds)."""

    async def go():
        t = make_transport(sshim)
        await t.connect()
        fail = sshim.tmp / "failctr"
        fail.write_text("1")
        os.environ["SSHIM_FAIL_FILE"] = str(fail)
        try:
            with pytest.raises(TransportConnectError):
                await t.run("true")
            assert not t.is_connected
            await t.connect()  # counter drained: reconnect succeeds
            assert t.is_connected
            assert (await t.run("echo back")).text_out().strip() == "back"
        finally:
            os.environ.pop("SSHIM_FAIL_FILE", None)
        await t.close()

    asyncio.run(go())


def test_env_prefix_injection_and_quoting(sshim):
    """Env vars ride an `export K=V && ` prefix through the login shell
    (GPU slot pinning depends on this); values with spaces, quotes and
    dollars must survive the ssh -- cmd re-parse."""

    async def go():
        t = make_transport(sshim)
        await t.connect()
        tricky = "a b'c\"d$HOME e;f&g"
        proc = await t.run(
            'printf "%s|%s" "$CSP_GPU_SLOT" "$TRICKY"',
            env={"CSP_GPU_SLOT": "5", "TRICKY": tricky},
        )
        assert proc.ok
        slot, got = proc.text_out().split("|", 1)
        assert slot == "5"
        assert got == tricky
        await t.close()

    asyncio.run(go())


def test_environment_is_sanitized(sshim, monkeypatch):
    """The shim gives the remote command a fresh minimal env, like sshd:
    dispatcher-process variables must NOT leak across the boundary."""
    monkeypatch.setenv("CSP_LEAK_CHECK", "leaked")

    async def go():
        t = make_transport(sshim)
        await t.connect()
        proc = await t.run('printf "%s" "${CSP_LEAK_CHECK:-clean}"')
        assert proc.text_out() == "clean"
        await t.close()

    asyncio.run(go())


def test_put_files_tar_batch_relative_and_get_file(sshim, tmp_path):
    """Batched tar-on-stdin staging into the remote home (replaces the
    reference's per-file scp sessions, reference ssh.py:360-361)."""

    async def go():
        t = make_transport(sshim)
        await t.connect()
        src1 = tmp_path / "one.bin"
        src2 = tmp_path / "two.txt"
        payload = os.urandom(256 * 1024)
        src1.write_bytes(payload)
        src2.write_text("text contents\n")
        await t.put_files(
            [
                (str(src1), ".cache/covalent/one.bin"),
                (str(src2), ".cache/covalent/sub/two.txt"),
            ]
        )
        assert (sshim.home / ".cache/covalent/one.bin").read_bytes() == payload
        assert (sshim.home / ".cache/covalent/sub/two.txt").read_text() == "text contents\n"

        back = tmp_path / "back.bin"
        await t.get_file(".cache/covalent/one.bin", str(back))
        assert back.read_bytes() == payload

        with pytest.raises(FileNotFoundError):
            await t.get_file(".cache/covalent/never-there", str(tmp_path / "x"))
        await t.close()

    asyncio.run(go())


def test_put_files_absolute_paths(sshim, tmp_path):
    async def go():
        t = make_transport(sshim)
        await t.connect()
        src = tmp_path / "abs.bin"
        src.write_bytes(b"absolute")
        dst = sshim.tmp / "absdest" / "abs.bin"
        await t.put_files([(str(src), str(dst))])
        assert dst.read_bytes() == b"absolute"
        await t.close()

    asyncio.run(go())


def test_stdin_binary_roundtrip(sshim):
    """Binary stdin (the fused dispatch's tar stream) passes through the
    client byte-exact, including NUL and sentinel-ish bytes."""

    async def go():
        t = make_transport(sshim)
        await t.connect()
        blob = bytes(range(256)) * 512 + b"--CSP-RESULT--" + os.urandom(1024)
        proc = await t.run("cat", input_data=blob)
        assert proc.ok
        assert proc.stdout == blob
        await t.close()

    asyncio.run(go())


def test_open_channel_frames(sshim):
    """A long-lived channel over the shim client: framed bytes flow both
    ways through one persistent remote process (`cat` echoes our frames
    verbatim — 8-byte length prefix + payload comes back identically)."""

    async def go():
        t = make_transport(sshim)
        await t.connect()
        ch = await t.open_channel("cat")
        for payload in (b"x", b"y" * 100_000, os.urandom(5 << 20)):
            await ch.send_frame(payload)
            got = await ch.recv_frame(timeout=30)
            assert bytes(got) == payload
        await ch.close()
        await t.close()

    asyncio.run(go())


def test_close_is_idempotent(sshim):
    async def go():
        t = make_transport(sshim)
        await t.connect()
        await t.close()
        await t.close()
        assert not t.is_connected

    asyncio.run(go())
